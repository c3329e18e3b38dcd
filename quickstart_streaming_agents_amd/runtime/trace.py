"""Tracing, metrics and failure detection for the streaming runtime.

The reference's only observability is `MAP['debug','true']` traces inside
managed Flink plus statement-status polling.  Here tracing
is first-class: per-stage spans with wall time and record counts, per-
episode agent traces (agents/runner.py debug traces plug in), pipeline
status in the reference's vocabulary (RUNNING / COMPLETED / FAILED /
DEGRADED — testing/helpers/flink_sql_helper.py:98-136), and the
retry-with-exponential-backoff helper the reference hand-rolls around
flaky externals (deploy.py:83-91, polling_helper.py:79-136).

GPU-side visibility comes from the kernels themselves (rocprofv3 sees
qsa_* kernels by name); this module covers the host runtime.
"""

from __future__ import annotations

import json
import time
from contextlib import contextmanager
from dataclasses import dataclass, field


@dataclass
class Span:
    stage: str
    t0: float
    dt: float = 0.0
    records_in: int = 0
    records_out: int = 0
    meta: dict = field(default_factory=dict)


class Tracer:
    """Per-pipeline stage spans + counters; cheap enough to leave on."""

    def __init__(self, pipeline: str = "", enabled: bool = True):
        self.pipeline = pipeline
        self.enabled = enabled
        self.spans: list[Span] = []
        self.counters: dict[str, float] = {}

    @contextmanager
    def stage(self, name: str, records_in: int = 0, **meta):
        if not self.enabled:
            yield None
            return
        sp = Span(name, time.perf_counter(), records_in=records_in,
                  meta=meta)
        try:
            yield sp
        finally:
            sp.dt = time.perf_counter() - sp.t0
            self.spans.append(sp)

    def count(self, name: str, value: float = 1.0) -> None:
        self.counters[name] = self.counters.get(name, 0.0) + value

    def summary(self) -> dict:
        by_stage: dict[str, dict] = {}
        for sp in self.spans:
            agg = by_stage.setdefault(sp.stage, {
                "calls": 0, "total_s": 0.0, "records_in": 0,
                "records_out": 0})
            agg["calls"] += 1
            agg["total_s"] += sp.dt
            agg["records_in"] += sp.records_in
            agg["records_out"] += sp.records_out
        return {"pipeline": self.pipeline, "stages": by_stage,
                "counters": dict(self.counters)}

    def dump_jsonl(self, path: str) -> None:
        with open(path, "w") as fh:
            for sp in self.spans:
                fh.write(json.dumps({
                    "pipeline": self.pipeline, "stage": sp.stage,
                    "t0": sp.t0, "dt": sp.dt, "in": sp.records_in,
                    "out": sp.records_out, **sp.meta}) + "\n")


# ---- pipeline status (statement-status parity) ----------------------------

RUNNING = "RUNNING"
COMPLETED = "COMPLETED"
FAILED = "FAILED"
DEGRADED = "DEGRADED"


class PipelineStatus:
    """Reference vocabulary: RUNNING is not enough — the E2E philosophy is
    'the agent must produce output' (testing/e2e/test_lab1.py:1-9), so
    DEGRADED flags a running pipeline whose output has stalled."""

    def __init__(self, name: str, stall_timeout_s: float = 60.0):
        self.name = name
        self.stall_timeout_s = stall_timeout_s
        self.state = RUNNING
        self.error: str | None = None
        self._last_output = time.monotonic()

    def record_output(self, n: int = 1) -> None:
        if n > 0:
            self._last_output = time.monotonic()

    def fail(self, err: str) -> None:
        self.state = FAILED
        self.error = err

    def complete(self) -> None:
        if self.state == RUNNING:
            self.state = COMPLETED

    @property
    def status(self) -> str:
        if self.state == RUNNING and \
                time.monotonic() - self._last_output > self.stall_timeout_s:
            return DEGRADED
        return self.state

    def is_terminal(self) -> bool:
        return self.state in (COMPLETED, FAILED)


def retry_with_backoff(fn, attempts: int = 5, base_delay_s: float = 0.1,
                       max_delay_s: float = 5.0, retry_on=Exception,
                       sleep=time.sleep):
    """Run fn() with exponential backoff; re-raises the last error."""
    delay = base_delay_s
    for i in range(attempts):
        try:
            return fn()
        except retry_on:
            if i == attempts - 1:
                raise
            sleep(delay)
            delay = min(delay * 2.0, max_delay_s)


def poll_until(predicate, timeout_s: float = 30.0, base_delay_s: float = 0.05,
               max_delay_s: float = 2.0, sleep=time.sleep,
               clock=time.monotonic):
    """Poll predicate() with exponential backoff until truthy or timeout;
    returns the truthy value or None (polling_helper.py:9-76 parity)."""
    deadline = clock() + timeout_s
    delay = base_delay_s
    while clock() < deadline:
        v = predicate()
        if v:
            return v
        sleep(min(delay, max(0.0, deadline - clock())))
        delay = min(delay * 2.0, max_delay_s)
    return None
