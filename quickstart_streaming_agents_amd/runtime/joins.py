"""Streaming joins with TTL state.

Two shapes the reference pipelines use:

- Keyed equi-join with state TTL: ``enriched_orders = orders JOIN customers
  JOIN products`` under ``SET 'sql.state-ttl' = '1 HOURS'``
  (LAB1-Walkthrough.md:119-131).
- Interval join: lab4's ``claims`` joined to the anomaly row where
  ``claim_timestamp BETWEEN window_time - 6h AND window_time``
  (LAB4-Walkthrough.md:231-237).

State lives in per-key dicts with lazy TTL eviction keyed off event time —
the single-node analog of Flink's keyed state backend.  The GPU hash-join
path (ops/hip) batches probe columns; this module is the engine-level
semantics + CPU reference.
"""

from __future__ import annotations

from typing import Any, Callable, Iterable


class TTLTable:
    """Latest-row-per-key state with event-time TTL eviction."""

    def __init__(self, key_fn: Callable[[dict], Any], ttl_ms: int | None = None):
        self.key_fn = key_fn
        self.ttl_ms = ttl_ms
        self._rows: dict[Any, tuple[int, dict]] = {}

    def upsert(self, row: dict, ts_ms: int) -> None:
        self._rows[self.key_fn(row)] = (ts_ms, row)

    def get(self, key: Any, now_ms: int) -> dict | None:
        item = self._rows.get(key)
        if item is None:
            return None
        ts, row = item
        if self.ttl_ms is not None and now_ms - ts > self.ttl_ms:
            del self._rows[key]
            return None
        return row

    def evict(self, now_ms: int) -> int:
        if self.ttl_ms is None:
            return 0
        dead = [k for k, (ts, _) in self._rows.items() if now_ms - ts > self.ttl_ms]
        for k in dead:
            del self._rows[k]
        return len(dead)

    def __len__(self) -> int:
        return len(self._rows)


def enrich_join(stream: Iterable[dict], ts_fn: Callable[[dict], int],
                dims: list[tuple[TTLTable, Callable[[dict], Any], str | None]],
                ) -> list[dict]:
    """Inner-join each stream row against dimension tables.

    dims: (table, probe_key_fn, prefix) — prefix namespaces the joined
    columns (None = merge raw, stream row wins on collision).  Rows missing
    any dimension are held back (inner-join semantics: they would emit later
    in Flink once the dimension arrives; bounded replays publish dims first,
    as the reference does: publish_lab1_data.py:377-385).
    """
    out = []
    for row in stream:
        now = ts_fn(row)
        merged = dict(row)
        ok = True
        for table, probe_fn, prefix in dims:
            hit = table.get(probe_fn(row), now)
            if hit is None:
                ok = False
                break
            if prefix is None:
                for k, v in hit.items():
                    merged.setdefault(k, v)
            else:
                for k, v in hit.items():
                    merged[f"{prefix}{k}"] = v
        if ok:
            out.append(merged)
    return out


def interval_join(left: Iterable[dict], right: list[dict],
                  left_ts: Callable[[dict], int], right_ts: Callable[[dict], int],
                  key_left: Callable[[dict], Any], key_right: Callable[[dict], Any],
                  lower_ms: int, upper_ms: int) -> list[dict]:
    """left x right where key matches and
    right_ts + lower <= left_ts <= right_ts + upper."""
    by_key: dict[Any, list[dict]] = {}
    for r in right:
        by_key.setdefault(key_right(r), []).append(r)
    out = []
    for l in left:
        lts = left_ts(l)
        for r in by_key.get(key_left(l), ()):
            rts = right_ts(r)
            if rts + lower_ms <= lts <= rts + upper_ms:
                merged = dict(r)
                merged.update(l)
                out.append(merged)
    return out
