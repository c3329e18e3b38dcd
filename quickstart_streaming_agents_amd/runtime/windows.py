"""Event-time tumbling windows + watermarks.

Semantics mirror the reference pipelines' TUMBLE usage
(LAB3-Walkthrough.md:99-133: 5-min windows per pickup_zone;
LAB4-Walkthrough.md:124-180: 6-h windows per city) and Flink's
window_time = window_end - 1ms rowtime convention.  A window closes when the
source watermark (max event ts - delay, per the table's WATERMARK clause)
passes its end.
"""

from __future__ import annotations

from dataclasses import dataclass, field
from typing import Any, Callable, Iterable


@dataclass
class WindowResult:
    key: Any
    window_start: int
    window_end: int
    rows: list[dict] = field(default_factory=list)

    @property
    def window_time(self) -> int:
        return self.window_end - 1


class Watermark:
    """Bounded-out-of-orderness watermark: max(event_ts) - delay."""

    def __init__(self, delay_ms: int = 5000):
        self.delay_ms = delay_ms
        self.max_ts = -(1 << 62)

    def observe(self, ts_ms: int) -> int:
        if ts_ms > self.max_ts:
            self.max_ts = ts_ms
        return self.current

    @property
    def current(self) -> int:
        return self.max_ts - self.delay_ms


class TumblingWindows:
    """Keyed tumbling window assigner/trigger.

    feed() buffers rows into (key, window) panes; advancing the watermark
    past a window end emits that pane exactly once, in (window_start, key)
    order for determinism.
    """

    def __init__(self, size_ms: int, key_fn: Callable[[dict], Any],
                 ts_fn: Callable[[dict], int], watermark_delay_ms: int = 5000):
        self.size_ms = size_ms
        self.key_fn = key_fn
        self.ts_fn = ts_fn
        self.wm = Watermark(watermark_delay_ms)
        self._panes: dict[tuple[Any, int], WindowResult] = {}
        self._late_dropped = 0

    def feed(self, rows: Iterable[dict]) -> list[WindowResult]:
        """Feed rows (any order within watermark bounds); return closed windows."""
        for row in rows:
            ts = self.ts_fn(row)
            start = (ts // self.size_ms) * self.size_ms
            if start + self.size_ms <= self.wm.current:
                self._late_dropped += 1  # late beyond watermark: dropped
                continue
            key = self.key_fn(row)
            pane = self._panes.get((key, start))
            if pane is None:
                pane = WindowResult(key, start, start + self.size_ms)
                self._panes[(key, start)] = pane
            pane.rows.append(row)
            self.wm.observe(ts)
        return self._drain()

    def _drain(self) -> list[WindowResult]:
        wm = self.wm.current
        ready = [k for k, p in self._panes.items() if p.window_end <= wm]
        ready.sort(key=lambda k: (self._panes[k].window_start, str(k[0])))
        return [self._panes.pop(k) for k in ready]

    @property
    def late_dropped(self) -> int:
        """Rows dropped for arriving beyond the watermark (the metric
        Flink exposes as numLateRecordsDropped)."""
        return self._late_dropped

    def flush(self) -> list[WindowResult]:
        """Close every remaining pane (end of bounded input)."""
        self.wm.observe(1 << 62)
        return self._drain()


def aggregate(panes: Iterable[WindowResult],
              aggs: dict[str, Callable[[list[dict]], Any]]) -> list[dict]:
    """Apply named aggregates per closed pane -> flat result rows."""
    out = []
    for p in panes:
        row = {
            "key": p.key,
            "window_start": p.window_start,
            "window_end": p.window_end,
            "window_time": p.window_time,
        }
        for name, fn in aggs.items():
            row[name] = fn(p.rows)
        out.append(row)
    return out
