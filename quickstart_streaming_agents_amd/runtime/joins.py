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


# ---------------------------------------------------------------------------
# GPU hash-join state (K8): columnar batches in HBM
# ---------------------------------------------------------------------------

def _key_hash(values) -> "object":
    """Stable 63-bit hashes for join keys (strings/ints), vectorized to an
    i64 torch tensor.  Deterministic across processes (blake2b), masked to
    63 bits so the table's EMPTY sentinel is unreachable."""
    import hashlib

    import torch
    out = torch.empty(len(values), dtype=torch.int64)
    for i, v in enumerate(values):
        b = v if isinstance(v, bytes) else str(v).encode()
        h = int.from_bytes(hashlib.blake2b(b, digest_size=8).digest(),
                          "little") & ((1 << 63) - 1)
        out[i] = h
    return out


class GpuTTLTable:
    """Latest-row-per-key join state resident in HBM.

    Build once per dimension batch (ops/hip/hash_join.hip: lock-free
    open-addressing insert, one atomicMax keeps the latest EVENT TIME per
    key — the deterministic order for a parallel batch, matching the
    reference's chronologically-replayed streams), probe with the stream
    batch under a TTL cutoff.  Rows stay on the host; the table stores
    row indices — the GPU does the keyed matching, the host does the
    dict assembly (cheap at emit width).

    Falls back to the same-semantics CPU dict when no GPU is present, so
    pipelines are testable anywhere (parity test in tests/test_gpu_kernels).
    """

    def __init__(self, rows: list[dict], key_col: str, ts_col: str | None,
                 ttl_ms: int | None = None, device: str | None = None):
        import torch
        self.rows = rows
        self.ts_col = ts_col
        self.ttl_ms = ttl_ms
        dev = device or ("cuda:0" if torch.cuda.is_available() else "cpu")
        keys = _key_hash([r[key_col] for r in rows]).to(dev)
        ts = torch.tensor(
            [int(r.get(ts_col, 0) or 0) if ts_col else 0 for r in rows],
            dtype=torch.int64, device=dev)
        self.device = dev
        self.max_ts = int(ts.max().item()) if rows else 0
        import quickstart_streaming_agents_amd.ops.dispatch as D
        self._D = D
        self.table = D.hash_build(keys, ts) if rows else None

    def probe(self, probe_keys: list, now_ms: int) -> list[dict | None]:
        """Latest un-expired dimension row per probe key (None = miss)."""
        hits = self.probe_rows(probe_keys)
        if self.ttl_ms is None:
            return hits
        cutoff = now_ms - self.ttl_ms
        return [h if h is not None and self._ts(h) >= cutoff else None
                for h in hits]

    def probe_rows(self, probe_keys: list) -> list[dict | None]:
        """Latest dimension row per key with NO TTL filter (one batched
        kernel probe); callers with per-row event times TTL-filter on the
        host from the hit row's own timestamp — identical semantics to a
        per-row-cutoff probe since the table keeps only the latest row."""
        if self.table is None or not probe_keys:
            return [None] * len(probe_keys)
        pk = _key_hash(probe_keys).to(self.device)
        rows_idx = self._D.hash_probe(self.table, pk, -(1 << 62))
        return [self.rows[i] if i >= 0 else None
                for i in rows_idx.cpu().tolist()]

    def _ts(self, row: dict) -> int:
        return int(row.get(self.ts_col, 0) or 0) if self.ts_col else 0


def enrich_join_columnar(stream: list[dict], ts_fn, dims: list[tuple],
                         device: str | None = None) -> list[dict]:
    """GPU-batched enrich_join: dims = (rows, key_col, ts_col, probe_col,
    ttl_ms).  Inner-join semantics identical to enrich_join (rows missing
    any dimension are held back)."""
    if not stream:
        return []
    tables = [(GpuTTLTable(rows, key_col, ts_col, ttl_ms, device), probe_col)
              for rows, key_col, ts_col, probe_col, ttl_ms in dims]
    keep = [dict(r) for r in stream]
    ok = [True] * len(stream)
    for table, probe_col in tables:
        idx = [i for i in range(len(stream)) if ok[i]]
        # ONE batched probe (no cutoff), then per-row TTL filtering on
        # the host against each stream row's own event time — identical
        # semantics to per-row-cutoff probes since the table keeps only
        # the latest row per key
        hits = table.probe_rows([stream[i][probe_col] for i in idx])
        ttl = table.ttl_ms
        for i, hit in zip(idx, hits):
            if hit is not None and ttl is not None and                     int(ts_fn(stream[i])) - table._ts(hit) > ttl:
                hit = None
            if hit is None:
                ok[i] = False
            else:
                for k, v in hit.items():
                    keep[i].setdefault(k, v)
    return [keep[i] for i in range(len(stream)) if ok[i]]
