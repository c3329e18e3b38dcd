"""Incremental (streaming) execution of the CTAS grammar.

`sql/exec.py` recomputes a CTAS from its source topics in one bounded
batch.  This module runs the same statements the way the reference's
Flink statements run — CONTINUOUSLY: each `advance()` consumes only the
records appended to the source topics since the last call, pushes them
through persistent operator state (tumbling-window panes + watermark,
per-key anomaly detectors, two-sided streaming-join buffers), and appends
only the NEW result rows to the sink topic.  Chained CTAS statements
compose through topics exactly like Flink jobs compose through Kafka.

State is checkpointable: `snapshot()`/`restore()` round-trip consumer
offsets, window panes, detector state and join buffers through the
JSON-serializable forms of runtime/checkpoint.py, so a pipeline can
resume mid-stream after a crash (the reference's recovery story is
replay-from-offset on Confluent Flink; SURVEY.md 2.5 elasticity row).

Join state is bounded by ``SET 'sql.state-ttl'`` (reference
LAB1-Walkthrough.md:119-120, LAB4:124): every buffered join row is
stamped with the statement's stream time (max event-time observed so
far, advanced by any source column the tables declare as event time —
their WATERMARK column or first TIMESTAMP column), and entries whose
stamp falls behind stream_time - ttl are evicted — the same
"state not updated for TTL" contract Flink's keyed-state TTL gives.
Statements with no `sql.state-ttl` keep state forever, as Flink does.
"""

from __future__ import annotations

import re
from typing import Any

from .catalog import analyze_select
from .exec import (SqlExecError, SqlExecutor, _extract_laterals,
                   _parse_joins, _parse_select_items, _Row, _split_bool,
                   _split_clauses)


class _TwoSidedJoin:
    """One streaming equi-join stage (+ residual predicates): buffers both
    sides, emits each matched pair exactly once (when the later side
    arrives).  Buffer entries are (stream_time_stamp, row); `evict()`
    drops entries whose stamp falls behind the TTL cutoff."""

    def __init__(self, ex: SqlExecutor, alias: str, cond: str):
        self.ex = ex
        self.alias = alias
        self.eq_left: list[str] = []     # expressions over the left row
        self.eq_right: list[str] = []    # column names on the right row
        self.residual: list[str] = []
        for t in _split_bool(cond):
            m = re.match(r"\s*([`\w.]+)\s*=\s*([`\w.]+)\s*$", t)
            if m:
                l, r = m.group(1).strip("`"), m.group(2).strip("`")
                if l.startswith(alias + "."):
                    l, r = r, l
                if r.startswith(alias + "."):
                    self.eq_left.append(l)
                    self.eq_right.append(r.split(".", 1)[1])
                    continue
            self.residual.append(t)
        self.left_buf: dict[tuple, list[tuple[int, _Row]]] = {}
        self.right_buf: dict[tuple, list[tuple[int, dict]]] = {}
        self.evicted = 0

    def _lkey(self, row: _Row) -> tuple:
        return tuple(self.ex.ev.eval(e, row) for e in self.eq_left)

    def _rkey(self, d: dict) -> tuple:
        return tuple(d.get(c) for c in self.eq_right)

    def _pair(self, row: _Row, d: dict) -> _Row | None:
        cand = row.child()
        cand.ns[self.alias] = d
        if all(self.ex.ev.pred(t, cand) for t in self.residual):
            return cand
        return None

    def on_left(self, rows: list[_Row], stamp: int = 0) -> list[_Row]:
        out = []
        for row in rows:
            k = self._lkey(row)
            self.left_buf.setdefault(k, []).append((stamp, row))
            for _, d in self.right_buf.get(k, ()):
                c = self._pair(row, d)
                if c is not None:
                    out.append(c)
        return out

    def on_right(self, dicts: list[dict], stamp: int = 0) -> list[_Row]:
        out = []
        for d in dicts:
            k = self._rkey(d)
            self.right_buf.setdefault(k, []).append((stamp, d))
            for _, row in self.left_buf.get(k, ()):
                c = self._pair(row, d)
                if c is not None:
                    out.append(c)
        return out

    def evict(self, cutoff: int) -> int:
        """Drop buffered entries stamped before `cutoff`; returns count."""
        n = 0
        for buf in (self.left_buf, self.right_buf):
            dead_keys = []
            for k, entries in buf.items():
                kept = [e for e in entries if e[0] >= cutoff]
                if len(kept) != len(entries):
                    n += len(entries) - len(kept)
                    if kept:
                        buf[k] = kept
                    else:
                        dead_keys.append(k)
            for k in dead_keys:
                del buf[k]
        self.evicted += n
        return n

    def size(self) -> int:
        return (sum(len(v) for v in self.left_buf.values())
                + sum(len(v) for v in self.right_buf.values()))

    def snapshot(self) -> dict:
        """JSON-serializable buffers (joins precede LATERAL stages in the
        grammar, so buffered rows are plain column dicts)."""
        return {"left": [[list(k), [[ts, r.ns] for ts, r in rows]]
                         for k, rows in self.left_buf.items()],
                "right": [[list(k), [[ts, d] for ts, d in ds]]
                          for k, ds in self.right_buf.items()],
                "evicted": self.evicted}

    def restore(self, snap: dict) -> None:
        self.left_buf = {tuple(k): [(ts, _Row(ns)) for ts, ns in rows]
                         for k, rows in snap.get("left", [])}
        self.right_buf = {tuple(k): [(ts, d) for ts, d in ds]
                          for k, ds in snap.get("right", [])}
        self.evicted = snap.get("evicted", 0)


class StreamingQuery:
    """One CTAS / INSERT..SELECT statement executed incrementally."""

    def __init__(self, ex: SqlExecutor, sink: str, select_sql: str):
        self.ex = ex
        self.sink = sink
        self.select_sql = select_sql
        self.info = analyze_select(select_sql)
        self.clauses = _split_clauses(select_sql)
        self.from_clause, self.laterals = _extract_laterals(
            self.clauses["from"])
        self.items = _parse_select_items(self.clauses["select"])
        self.limit = (int(self.clauses["limit"].split()[0])
                      if self.clauses["limit"] else None)
        self.emitted = 0
        self._offsets: dict[str, dict[int, int]] = {}  # table -> {part -> off}
        self._consumers: dict[str, Any] = {}

        if self.info.tumble:
            from ..runtime.anomaly import AnomalyDetector
            from ..runtime.windows import TumblingWindows
            tum = self.info.tumble
            mo = re.search(r"OVER\s*\(\s*PARTITION\s+BY\s+([`\w]+)",
                           select_sql, re.IGNORECASE)
            mg = re.search(r"GROUP\s+BY\s+([`\w]+)", select_sql,
                           re.IGNORECASE)
            self.key_col = (mo or mg).group(1).strip("`") if (mo or mg) \
                else None
            if self.key_col is None:
                raise SqlExecError("TUMBLE without a key column")
            self.aggs = {}
            for expr, alias in self.items:
                m = re.match(r"(COUNT|SUM|AVG)\s*\(\s*(.*)\s*\)$",
                             expr.strip(), re.IGNORECASE | re.DOTALL)
                if m:
                    self.aggs[alias] = (m.group(1).upper(),
                                        m.group(2).strip())
            self.windows = TumblingWindows(
                tum["window_ms"], key_fn=lambda r: r[self.key_col],
                ts_fn=lambda r: r[tum["ts_col"]],
                watermark_delay_ms=ex.watermark_delay_ms(tum["table"]))
            self.detector = None
            self.anom_alias = "anomaly"
            self.anom_value_expr = None
            if self.info.anomaly:
                from . import parse as P
                ma = re.search(r"ML_DETECT_ANOMALIES\s*\(", select_sql,
                               re.IGNORECASE)
                close = P._find_matching_paren(select_sql, ma.end() - 1)
                self.anom_value_expr = P._split_top(
                    select_sql[ma.end():close])[0]
                mal = re.search(r"\)\s*AS\s+(\w+)", select_sql[close:],
                                re.IGNORECASE)
                if mal:
                    self.anom_alias = mal.group(1)
                self.detector = AnomalyDetector.from_json_params(
                    self.info.anomaly[0])
            self.stream_table = tum["table"]
            self.stream_alias = tum["table"]
            self.joins = []
        else:
            tables = _parse_joins(self.from_clause)
            if not tables:
                raise SqlExecError(f"empty FROM in {sink}")
            self.stream_table, self.stream_alias, _ = tables[0]
            self.joins = [
                (name, _TwoSidedJoin(ex, alias, cond))
                for name, alias, cond in tables[1:]]

        # ---- state TTL (SET 'sql.state-ttl'; LAB1-Walkthrough.md:119-120)
        self.ttl_ms = ex.catalog.state_ttl_ms()
        self._stream_time = -(1 << 62)   # max event ts observed so far
        self._ts_cols = {t: self._event_time_col(t)
                         for t in [self.stream_table]
                         + [name for name, _ in self.joins]}

    def _event_time_col(self, table: str) -> str | None:
        """The column that advances stream time for TTL: the table's
        WATERMARK column if declared, else its first TIMESTAMP column."""
        t = self.ex.catalog.tables.get(table)
        if t is None:
            return None
        if t.watermark:
            return t.watermark[0]
        for c in t.columns:
            if c.type.upper().startswith("TIMESTAMP"):
                return c.name
        return None

    def _observe(self, table: str, dicts: list[dict]) -> None:
        col = self._ts_cols.get(table)
        if col is None:
            return
        for d in dicts:
            v = d.get(col)
            if isinstance(v, (int, float)) and v > self._stream_time:
                self._stream_time = int(v)

    # -- incremental sources ------------------------------------------------
    def _new_source_rows(self, table: str) -> list[dict]:
        t = self.ex.catalog.tables.get(table)
        schema = self.ex.schemas.get(table)
        if schema is not None and (t is None or t.as_select is None):
            from ..wire.topics import AvroConsumer
            c = self._consumers.get(table)
            if c is None:
                c = self._consumers[table] = AvroConsumer(
                    self.ex.broker, table, schema)
            return [v for _, v in c.poll()]
        topic = self.ex.broker.topics.get(table)
        if topic is None:
            return []
        # per-partition offsets (like AvroConsumer.poll): a late-timestamp
        # append must be delivered exactly once, which a slice of the
        # globally timestamp-sorted read_all() cannot guarantee
        offs = self._offsets.setdefault(table, {})
        out = []
        for pi, p in enumerate(topic.partitions):
            start = offs.get(pi, 0)
            recs = p.read(start)
            offs[pi] = start + len(recs)
            for rec in recs:
                v = rec.value
                out.append(v if isinstance(v, dict) else {"value": v})
        return out

    # -- stages -------------------------------------------------------------
    def _window_close(self, panes) -> list[_Row]:
        rows = []
        for p in panes:
            cols = {self.key_col: p.key, "window_start": p.window_start,
                    "window_end": p.window_end,
                    "window_time": p.window_time}
            for alias, (fn, arg) in self.aggs.items():
                if fn == "COUNT":
                    cols[alias] = len(p.rows)
                else:
                    vals = [float(self.ex.ev.eval(arg, _Row({"_": rr})))
                            for rr in p.rows]
                    cols[alias] = (sum(vals) if fn == "SUM"
                                   else sum(vals) / max(len(vals), 1))
            row = _Row({"_w": cols})
            if self.detector is not None:
                res = self.detector.update(
                    p.key, float(self.ex.ev.eval(self.anom_value_expr, row)))
                row.ns[self.anom_alias] = {
                    "forecast_value": res.forecast_value,
                    "upper_bound": res.upper_bound,
                    "lower_bound": res.lower_bound,
                    "is_anomaly": res.is_anomaly,
                }
            rows.append(row)
        return rows

    def _finish_rows(self, rows: list[_Row]) -> list[dict]:
        for lat in self.laterals:
            if not rows:
                break
            rows = self.ex._apply_lateral(lat, rows, self.items,
                                          self.select_sql)
        for key in ("where", "having"):
            cond = self.clauses[key]
            if cond:
                rows = [r for r in rows if self.ex.ev.pred(cond, r)]
        if self.limit is not None:
            room = max(0, self.limit - self.emitted)
            rows = rows[:room]
        out = [self.ex._project(self.items, r) for r in rows]
        self.emitted += len(out)
        topic = self.ex.broker.create_topic(self.sink)
        for row in out:
            topic.append(row, partition=0)
        return out

    # -- public -------------------------------------------------------------
    def advance(self) -> list[dict]:
        """Consume newly-arrived source records, emit new result rows."""
        if self.limit is not None and self.emitted >= self.limit:
            return []
        if self.info.tumble:
            new = self._new_source_rows(self.stream_table)
            panes = self.windows.feed(new)
            panes.sort(key=lambda p: (p.window_start, str(p.key)))
            return self._finish_rows(self._window_close(panes))
        new_left = self._new_source_rows(self.stream_table)
        self._observe(self.stream_table, new_left)
        new_right: dict[str, list[dict]] = {}
        for name, _ in self.joins:
            if name not in new_right:   # self-joins share one fetch
                new_right[name] = self._new_source_rows(name)
                self._observe(name, new_right[name])
        stamp = self._stream_time
        rows = [_Row({self.stream_alias: d}) for d in new_left]
        finished: list[_Row] = []
        for i, (name, stage) in enumerate(self.joins):
            # new right-side rows first: they match already-buffered lefts.
            # The cascade through later stages fully joins them, so they go
            # straight to `finished` — pushing them back through `rows`
            # would buffer and emit each matched pair twice.
            from_right = stage.on_right(new_right[name], stamp)
            for _, later in self.joins[i + 1:]:
                from_right = later.on_left(from_right, stamp)
            finished.extend(from_right)
            rows = stage.on_left(rows, stamp)
        if self.ttl_ms is not None and self.joins:
            cutoff = self._stream_time - self.ttl_ms
            for _, stage in self.joins:
                stage.evict(cutoff)
        return self._finish_rows(rows + finished)

    def stats(self) -> dict:
        """Operator-level counters for monitoring (rows emitted, join
        buffer sizes, late-dropped window rows)."""
        out = {"sink": self.sink, "emitted": self.emitted}
        if self.info.tumble:
            out["late_dropped"] = self.windows.late_dropped
            out["open_panes"] = len(self.windows._panes)
        if self.joins:
            out["join_buffered"] = sum(st.size() for _, st in self.joins)
            out["join_evicted"] = sum(st.evicted for _, st in self.joins)
        return out

    def flush(self) -> list[dict]:
        """Bounded-input end: close every remaining window pane."""
        if not self.info.tumble:
            return []
        panes = self.windows.flush()
        panes.sort(key=lambda p: (p.window_start, str(p.key)))
        return self._finish_rows(self._window_close(panes))

    # -- checkpoint ---------------------------------------------------------
    def snapshot(self) -> dict:
        from ..runtime.checkpoint import snapshot_anomaly, snapshot_windows
        snap: dict = {"offsets": {t: {str(p): o for p, o in offs.items()}
                                  for t, offs in self._offsets.items()},
                      "stream_time": self._stream_time,
                      "consumer_offsets": {
                          t: c.offsets() if hasattr(c, "offsets")
                          else getattr(c, "_offsets", None)
                          for t, c in self._consumers.items()},
                      "emitted": self.emitted}
        if self.info.tumble:
            snap["windows"] = snapshot_windows(self.windows)
            if self.detector is not None:
                snap["anomaly"] = snapshot_anomaly(self.detector)
        if self.joins:
            snap["joins"] = [s.snapshot() for _, s in self.joins]
        return snap

    def restore(self, snap: dict) -> None:
        from ..runtime.checkpoint import restore_anomaly, restore_windows
        self._offsets = {}
        for t, offs in snap.get("offsets", {}).items():
            if isinstance(offs, dict):
                self._offsets[t] = {int(p): o for p, o in offs.items()}
            else:   # legacy count form (single-partition topics)
                self._offsets[t] = {0: int(offs)}
        self._stream_time = snap.get("stream_time", -(1 << 62))
        self.emitted = snap.get("emitted", 0)
        for t, offs in (snap.get("consumer_offsets") or {}).items():
            if offs is None:
                continue
            c = self._consumers.get(t)
            if c is None:
                from ..wire.topics import AvroConsumer
                c = self._consumers[t] = AvroConsumer(
                    self.ex.broker, t, self.ex.schemas[t])
            c._offsets = {int(k): v for k, v in offs.items()}
        if self.info.tumble:
            restore_windows(self.windows, snap["windows"])
            if self.detector is not None and "anomaly" in snap:
                restore_anomaly(self.detector, snap["anomaly"])
        for (_, stage), s in zip(self.joins, snap.get("joins", [])):
            stage.restore(s)


class StreamingPipeline:
    """All of a catalog's CTAS/INSERT statements as one streaming job
    graph, advanced in dependency order (upstream statements first, so an
    advance cascades through the chain within one call)."""

    def __init__(self, ex: SqlExecutor, final_tables: list[str] | None = None):
        self.ex = ex
        stmts: list[tuple[str, str]] = []
        for ins in ex.catalog.inserts:
            if ins.select:
                stmts.append((ins.table, ins.select))
        for name, t in ex.catalog.tables.items():
            if t.as_select:
                stmts.append((name, t.as_select))
        order = self._topo(stmts)
        self.queries = [StreamingQuery(ex, sink, sql)
                        for sink, sql in order]
        self.by_sink = {q.sink: q for q in self.queries}

    @staticmethod
    def _topo(stmts):
        deps = {}
        for sink, sql in stmts:
            info = analyze_select(sql)
            srcs = set(info.source_tables)
            if info.tumble:
                srcs.add(info.tumble["table"])
            deps[sink] = (srcs, sql)
        out, done = [], set()
        def visit(sink, stack=()):
            if sink in done or sink not in deps:
                return
            if sink in stack:
                raise SqlExecError(f"CTAS cycle at {sink}")
            srcs, sql = deps[sink]
            for s in srcs:
                visit(s, stack + (sink,))
            done.add(sink)
            out.append((sink, sql))
        for sink in deps:
            visit(sink)
        return out

    def advance(self) -> dict[str, list[dict]]:
        return {q.sink: q.advance() for q in self.queries}

    def finish(self) -> dict[str, list[dict]]:
        """Bounded end: flush windows, then cascade until quiescent."""
        out: dict[str, list[dict]] = {q.sink: [] for q in self.queries}
        for q in self.queries:
            out[q.sink] += q.flush()
            out[q.sink] += q.advance()
        # cascade downstream of late window closures
        for _ in range(len(self.queries)):
            moved = False
            for q in self.queries:
                new = q.advance()
                if new:
                    moved = True
                    out[q.sink] += new
            if not moved:
                break
        return out

    def stats(self) -> list[dict]:
        return [q.stats() for q in self.queries]

    def snapshot(self) -> dict:
        return {q.sink: q.snapshot() for q in self.queries}

    def restore(self, snap: dict) -> None:
        for sink, s in snap.items():
            if sink in self.by_sink:
                self.by_sink[sink].restore(s)

    # -- durable form (runtime/checkpoint.py: torn-write-safe files) -----
    def checkpoint(self, root: str, pipeline: str = "sql",
                   shard: int = 0) -> int:
        """Persist the pipeline state to disk; returns the checkpoint id."""
        from ..runtime.checkpoint import CheckpointStore, PipelineState
        store = CheckpointStore(root, pipeline, shard)
        return store.save(PipelineState(operator=self.snapshot()))

    def resume(self, root: str, pipeline: str = "sql",
               shard: int = 0) -> bool:
        """Restore the latest on-disk checkpoint; False if none exists."""
        from ..runtime.checkpoint import CheckpointStore
        state = CheckpointStore(root, pipeline, shard).load()
        if state is None:
            return False
        self.restore(state.operator)
        return True
