"""Generic executor for the CTAS/INSERT subset grammar.

The four lab pipelines (labs/pipelines.py) are hand-written, operator-fused
implementations of the lab SQL.  This module executes the SAME statement
grammar *generically*: a user can write new CREATE TABLE ... AS SELECT
statements in the reference's Flink-SQL dialect —
streaming joins, TUMBLE + ML_DETECT_ANOMALIES, LATERAL ML_PREDICT /
VECTOR_SEARCH_AGG / AI_RUN_AGENT, REGEXP_EXTRACT projections — and run
them against broker topics without writing Python.  Execution is
vectorized stage-by-stage so LLM/embedding calls batch across rows
(the same batching discipline the hand-written pipelines use), which is
what keeps the GPU engine efficient under this executor too.

Supported surface (the grammar of labs/sql/*.sql):
  FROM t [alias] [JOIN t2 [a2] ON eq [AND cond]...]...
  FROM TABLE(TUMBLE(TABLE t, DESCRIPTOR(ts), INTERVAL 'n' U))
    with COUNT(*)/SUM(x)/AVG(x) select aggregates + GROUP BY
  ML_DETECT_ANOMALIES(value, ts, JSON_OBJECT(...)) OVER (PARTITION BY k
    ORDER BY window_time ...) AS alias
  [CROSS JOIN] LATERAL TABLE(ML_PREDICT('model', expr)) [AS alias]
  [CROSS JOIN] LATERAL TABLE(VECTOR_SEARCH_AGG(tbl, DESCRIPTOR(c), q, k))
    AS alias            -- alias.chunk/score/document_id => row per hit;
                        -- alias.chunk1..chunkK => numbered single row
  LATERAL TABLE(AI_RUN_AGENT('agent', prompt[, key][, MAP[...]]))
    AS alias(col, ...)
  WHERE / HAVING conjunctions of comparisons, <>, BETWEEN, bare booleans
  ORDER BY col [ASC|DESC][, ...]; LIMIT n
  expressions: literals, refs, CONCAT, CAST, COALESCE, UPPER/LOWER,
  REGEXP_EXTRACT, CASE WHEN, MAP[...] literals, +/- INTERVAL 'n' U
  scalar smoke forms: SELECT ML_PREDICT('m', 'q') / AI_TOOL_INVOKE(...)
  without FROM (LAB1-Walkthrough.md:66-92)

Unbound identifiers (the reference inlines giant CONCAT prompt
expressions; we hoist them into named *bindings*) resolve through the
`bindings` map: name -> fn(resolver) -> value, letting deployments supply
prompt builders (user_prompt, surge_query, investigation_prompt, ...)
without widening the SQL grammar.
"""

from __future__ import annotations

import re
from typing import Callable

from . import parse as P
from .catalog import Catalog, analyze_select

_NUM_RE = re.compile(r"^-?\d+(\.\d+)?$")
_IDENT_RE = re.compile(r"^[`\w.]+$")


class SqlExecError(ValueError):
    pass


class _Row:
    """One working row: ordered namespaces (FROM alias -> column dict)."""

    __slots__ = ("ns", "last_embedding")

    def __init__(self, ns: dict[str, dict] | None = None):
        self.ns: dict[str, dict] = dict(ns or {})
        self.last_embedding = None

    def child(self) -> "_Row":
        r = _Row(self.ns)
        r.last_embedding = self.last_embedding
        return r

    def resolve(self, name: str):
        name = name.strip("`")
        if "." in name:
            alias, col = name.split(".", 1)
            d = self.ns.get(alias)
            if d is not None:
                if col in d:
                    return d[col]
                raise KeyError(name)
            # alias gone (e.g. after GROUP BY collapsed namespaces):
            # fall through to a bare-column lookup
            name = col
        for d in self.ns.values():
            if name in d:
                return d[name]
        raise KeyError(name)


class Evaluator:
    """Scalar-expression evaluator over a _Row + bindings."""

    def __init__(self, bindings: dict[str, Callable] | None = None):
        self.bindings = bindings or {}
        self.fns: dict[str, Callable] = {}   # extension functions
                                             # (executor registers
                                             # ML_PREDICT / AI_TOOL_INVOKE
                                             # scalar forms)

    # -- helpers -----------------------------------------------------------
    @staticmethod
    def _unquote(tok: str) -> str:
        return tok[1:-1].replace("''", "'")

    def _interval_split(self, expr: str):
        """a +/- INTERVAL 'n' UNIT -> (a, signed ms) or None."""
        m = re.search(r"([+-])\s*INTERVAL\s+'(\d+)'\s+(\w+)\s*$", expr,
                      re.IGNORECASE)
        if not m:
            return None
        from .catalog import _interval_ms
        ms = _interval_ms(m.group(2), m.group(3))
        return expr[:m.start()].strip(), (ms if m.group(1) == "+" else -ms)

    # -- entry -------------------------------------------------------------
    def eval(self, expr: str, row: _Row):
        expr = expr.strip()
        iv = self._interval_split(expr)
        if iv is not None:
            base, ms = iv
            return int(self.eval(base, row)) + ms
        if expr.startswith("'"):
            return self._unquote(expr)
        if _NUM_RE.match(expr):
            return float(expr) if "." in expr else int(expr)
        up = expr.upper()
        if up in ("TRUE", "FALSE"):
            return up == "TRUE"
        if up.startswith("MAP[") and expr.endswith("]"):
            # MAP['k','v',...] literal -> dict (MAP[] -> {})
            inner = expr[4:-1].strip()
            if not inner:
                return {}
            vals = [self.eval(a, row) for a in P._split_top(inner)]
            return dict(zip(vals[0::2], vals[1::2]))
        if up.startswith("CASE") and up.endswith("END"):
            return self._case(expr, row)
        m = re.match(r"(\w+)\s*\(", expr)
        if m and expr.endswith(")"):
            close = P._find_matching_paren(expr, m.end() - 1)
            if close == len(expr) - 1:
                fn = m.group(1).upper()
                args = P._split_top(expr[m.end():close])
                return self._call(fn, args, row, expr)
        if _IDENT_RE.match(expr):
            try:
                return row.resolve(expr)
            except KeyError:
                name = expr.strip("`").split(".")[-1]
                if name in self.bindings:
                    return self.bindings[name](lambda n: row.resolve(n))
                # reference SQL renames LATERAL embedding outputs freely
                # (narrative_embedding, rad.embedding, ...): any unresolved
                # *embedding name falls back to the row's last ML_PREDICT
                # embedding output
                if name.endswith("embedding") and \
                        row.last_embedding is not None:
                    return row.last_embedding
                raise SqlExecError(f"unbound identifier {expr!r} "
                                   f"(add a binding?)")
        raise SqlExecError(f"unsupported expression {expr!r}")

    def _case(self, expr: str, row: _Row):
        """CASE WHEN cond THEN v [WHEN ...] [ELSE v] END (searched form)."""
        body = expr.strip()[4:].strip()
        if body.upper().endswith("END"):
            body = body[:-3].strip()
        # split on top-level WHEN/ELSE keywords
        tokens = re.split(r"\b(WHEN|THEN|ELSE)\b", body,
                          flags=re.IGNORECASE)
        # tokens like ['', 'WHEN', cond, 'THEN', val, 'WHEN', ...]
        i = 0
        default = None
        while i < len(tokens):
            kw = tokens[i].strip().upper()
            if kw == "WHEN":
                cond, val = tokens[i + 1], tokens[i + 3]
                assert tokens[i + 2].strip().upper() == "THEN", expr
                if self.pred(cond, row):
                    return self.eval(val, row)
                i += 4
            elif kw == "ELSE":
                default = tokens[i + 1]
                i += 2
            else:
                i += 1
        return self.eval(default, row) if default is not None else None

    def _call(self, fn: str, args: list[str], row: _Row, expr: str):
        if fn == "CONCAT":
            return "".join(str(self.eval(a, row)) for a in args)
        if fn == "CAST":
            inner, _, ty = args[0].rpartition(" AS ")
            v = self.eval(inner.strip(), row)
            ty = ty.strip().upper()
            if ty.startswith(("DOUBLE", "FLOAT", "DECIMAL")):
                return float(v)
            if ty.startswith(("INT", "BIGINT")):
                return int(float(v))
            return str(v)
        if fn == "COALESCE":
            for a in args:
                v = self.eval(a, row)
                if v is not None:
                    return v
            return None
        if fn in ("UPPER", "LOWER"):
            v = str(self.eval(args[0], row))
            return v.upper() if fn == "UPPER" else v.lower()
        if fn == "REGEXP_EXTRACT":
            from ..agents.parse import regexp_extract
            subject = self.eval(args[0], row)
            pattern = self._unquote(args[1].strip())
            group = int(args[2]) if len(args) > 2 else 1
            return regexp_extract(str(subject or ""), pattern, group)
        if fn in self.fns:
            return self.fns[fn](args, row)
        raise SqlExecError(f"unsupported function in {expr!r}")

    # -- boolean predicates -------------------------------------------------
    def pred(self, cond: str, row: _Row) -> bool:
        for term in _split_bool(cond):
            if not self._term(term, row):
                return False
        return True

    def _term(self, term: str, row: _Row) -> bool:
        term = term.strip()
        m = re.match(r"(.+?)\s+BETWEEN\s+(.+?)\s+AND\s+(.+)$", term,
                     re.IGNORECASE | re.DOTALL)
        if m:
            v = self.eval(m.group(1), row)
            return self.eval(m.group(2), row) <= v <= self.eval(m.group(3), row)
        for op in ("<>", "!=", ">=", "<=", "=", ">", "<"):
            i = _find_top_level(term, op)
            if i >= 0:
                l = self.eval(term[:i], row)
                r = self.eval(term[i + len(op):], row)
                if op in ("<>", "!="):
                    return l != r
                if op == "=":
                    return l == r
                l, r = float(l), float(r)
                return {"<": l < r, ">": l > r,
                        "<=": l <= r, ">=": l >= r}[op]
        return bool(self.eval(term, row))    # bare boolean ref


def _split_bool(cond: str) -> list[str]:
    """Split on top-level AND (outside parens/strings)."""
    parts, cur, depth, in_str = [], [], 0, False
    i, n = 0, len(cond)
    while i < n:
        ch = cond[i]
        if in_str:
            cur.append(ch)
            if ch == "'":
                if i + 1 < n and cond[i + 1] == "'":
                    cur.append("'")
                    i += 1
                else:
                    in_str = False
        elif ch == "'":
            in_str = True
            cur.append(ch)
        elif ch == "(":
            depth += 1
            cur.append(ch)
        elif ch == ")":
            depth -= 1
            cur.append(ch)
        elif depth == 0 and cond[i:i + 5].upper() == " AND ":
            parts.append("".join(cur))
            cur = []
            i += 4
        else:
            cur.append(ch)
        i += 1
    parts.append("".join(cur))
    # re-join the AND that belongs to a BETWEEN ... AND ... range
    out: list[str] = []
    for p in parts:
        if out and len(re.findall(r"\bBETWEEN\b", out[-1], re.IGNORECASE)) \
                > len(re.findall(r"\bAND\b", out[-1], re.IGNORECASE)):
            out[-1] += " AND " + p
        else:
            out.append(p)
    return [p.strip() for p in out if p.strip()]


def _find_top_level(s: str, needle: str) -> int:
    depth, in_str = 0, False
    i, n = 0, len(s)
    while i < n:
        ch = s[i]
        if in_str:
            if ch == "'":
                if i + 1 < n and s[i + 1] == "'":
                    i += 1
                else:
                    in_str = False
        elif ch == "'":
            in_str = True
        elif ch == "(":
            depth += 1
        elif ch == ")":
            depth -= 1
        elif depth == 0 and s.startswith(needle, i):
            # avoid matching '<' inside '<>' etc: caller orders ops longest-first
            return i
        i += 1
    return -1


# ---------------------------------------------------------------------------
# SELECT clause decomposition
# ---------------------------------------------------------------------------

_CLAUSE_RE = re.compile(
    r"\bSELECT\b(?P<select>.*?)\bFROM\b(?P<rest>.*)$",
    re.IGNORECASE | re.DOTALL)


_CLAUSE_KWS = ("WHERE", "GROUP BY", "HAVING", "ORDER BY", "LIMIT")


def _kw_top(s: str, kws=_CLAUSE_KWS) -> tuple[int, int, str] | None:
    """First top-level (outside parens/strings) clause keyword:
    (start, end, keyword) or None."""
    depth, in_str = 0, False
    i, n = 0, len(s)
    while i < n:
        ch = s[i]
        if in_str:
            if ch == "'":
                if i + 1 < n and s[i + 1] == "'":
                    i += 1
                else:
                    in_str = False
        elif ch == "'":
            in_str = True
        elif ch == "(":
            depth += 1
        elif ch == ")":
            depth -= 1
        elif depth == 0 and (i == 0 or not s[i - 1].isalnum()):
            for kw in kws:
                if s[i:i + len(kw)].upper() == kw and \
                        (i + len(kw) >= n or not s[i + len(kw)].isalnum()):
                    return i, i + len(kw), kw
        i += 1
    return None


def _split_clauses(select_sql: str) -> dict:
    m = _CLAUSE_RE.search(select_sql)
    if not m:
        ms = re.match(r"\s*SELECT\b(.*)$", select_sql,
                      re.IGNORECASE | re.DOTALL)
        if not ms:
            raise SqlExecError("no SELECT found")
        return {"select": ms.group(1).strip().rstrip(";").strip(),
                "from": None, "where": None, "having": None,
                "group_by": None, "order_by": None, "limit": None}
    rest = m.group("rest")
    out = {"select": m.group("select").strip(), "where": None,
           "having": None, "group_by": None, "order_by": None,
           "limit": None}
    first = _kw_top(rest)
    out["from"] = (rest[:first[0]] if first else rest) \
        .strip().rstrip(";").strip()
    while first:
        start, end, kw = first
        tail = rest[end:]
        nxt = _kw_top(tail)
        body = (tail[:nxt[0]] if nxt else tail).strip().rstrip(";").strip()
        out[kw.lower().replace(" ", "_")] = body
        rest = tail
        first = nxt
    return out


def _parse_select_items(select: str) -> list[tuple[str, str]]:
    """-> [(expr, alias)]; alias defaults to the ref's last path part."""
    items = []
    for item in P._split_top(select):
        item = item.strip()
        m = re.search(r"\s+AS\s+([`\w]+)\s*$", item, re.IGNORECASE)
        if m and _balanced(item[:m.start()]):
            expr, alias = item[:m.start()].strip(), m.group(1).strip("`")
        else:
            expr, alias = item, item.strip("`").split(".")[-1]
        items.append((expr, alias))
    return items


def _balanced(s: str) -> bool:
    depth, in_str = 0, False
    i = 0
    while i < len(s):
        ch = s[i]
        if in_str:
            if ch == "'":
                if i + 1 < len(s) and s[i + 1] == "'":
                    i += 1
                else:
                    in_str = False
        elif ch == "'":
            in_str = True
        elif ch == "(":
            depth += 1
        elif ch == ")":
            depth -= 1
        i += 1
    return depth == 0 and not in_str


_LATERAL_RE = re.compile(
    r"(?:CROSS\s+JOIN\s+)?LATERAL\s+TABLE\s*\(", re.IGNORECASE)


def _extract_laterals(from_clause: str):
    """Pull LATERAL TABLE(...) [AS alias[(cols)]] items out of FROM."""
    laterals = []
    s = from_clause
    while True:
        m = _LATERAL_RE.search(s)
        if not m:
            break
        close = P._find_matching_paren(s, m.end() - 1)
        call = s[m.end():close].strip()
        tail = s[close + 1:]
        alias, cols = None, None
        ma = re.match(r"\s*AS\s+(\w+)\s*(\(([^)]*)\))?", tail,
                      re.IGNORECASE)
        end = close + 1
        if ma:
            alias = ma.group(1)
            if ma.group(3):
                cols = [c.strip() for c in ma.group(3).split(",")]
            end = close + 1 + ma.end()
        laterals.append({"call": call, "alias": alias, "cols": cols})
        s = s[:m.start()] + s[end:]
    s = re.sub(r",\s*(,|$)", r"\1", s).strip().rstrip(",").strip()
    return s, laterals


def _parse_joins(from_clause: str):
    """'t1 a JOIN t2 b ON ... [JOIN t3 c ON ...]' or 't1 a, t2 b'."""
    parts = re.split(r"\bJOIN\b", from_clause, flags=re.IGNORECASE)
    first = parts[0].strip().rstrip(",").strip()
    tables = []
    for item in P._split_top(first):
        toks = item.strip().split()
        if not toks:
            continue
        name = toks[0].strip("`")
        alias = toks[1].strip("`") if len(toks) > 1 else name
        tables.append((name, alias, None))
    for part in parts[1:]:
        m = re.match(r"\s*([`\w.-]+)\s+(\w+\s+)?ON\s+(.*)$", part.strip(),
                     re.IGNORECASE | re.DOTALL)
        if not m:
            raise SqlExecError(f"cannot parse JOIN clause: {part!r}")
        name = m.group(1).strip("`")
        alias = (m.group(2) or name).strip().strip("`")
        tables.append((name, alias, m.group(3).strip().rstrip(",").strip()))
    return tables


# ---------------------------------------------------------------------------
# Executor
# ---------------------------------------------------------------------------

class SqlExecutor:
    """Executes catalog CTAS/INSERT statements over broker topics.

    Parameters
    ----------
    catalog:    sql.catalog.Catalog with the DDL applied
    broker:     wire broker (topic(name).read_all() / create_topic)
    schemas:    topic name -> Avro schema (for decoding source topics)
    embedder:   .embed/.embed_batch (K1)
    indexes:    vector table name -> VectorIndex (K2)
    llm_batch:  (prompts, max_tokens) -> texts (K4)
    tool_fn:    (name, args) -> result for agent episodes (K5/K6)
    bindings:   unbound identifier -> fn(resolver) -> value
    agent_policies: agent name -> fn(resolver) -> episode policy
    """

    def __init__(self, catalog: Catalog, broker, schemas: dict | None = None,
                 embedder=None, indexes: dict | None = None,
                 llm_batch=None, tool_fn=None,
                 bindings: dict[str, Callable] | None = None,
                 agent_policies: dict[str, Callable] | None = None,
                 max_new_tokens: int = 96, tracer=None,
                 tool_schemas: dict | None = None):
        self.catalog = catalog
        self.broker = broker
        self.schemas = schemas or {}
        self.embedder = embedder
        self.indexes = indexes or {}
        self.llm_batch = llm_batch
        self.tool_fn = tool_fn
        self.ev = Evaluator(bindings)
        self.agent_policies = agent_policies or {}
        # MCP tool schemas (tools/list): agents WITHOUT an installed
        # scripted policy run model-driven via grammar-constrained
        # decoding (models/grammar.py), like the bench and serving paths
        self.tool_schemas = tool_schemas
        self.max_new_tokens = max_new_tokens
        if tracer is None:
            from ..runtime.trace import Tracer
            tracer = Tracer("sql", enabled=False)
        self.tracer = tracer
        self.ev.fns["ML_PREDICT"] = self._fn_ml_predict
        self.ev.fns["AI_TOOL_INVOKE"] = self._fn_tool_invoke
        self._cache: dict[str, list[dict]] = {}

    # -- row sources --------------------------------------------------------
    def table_rows(self, name: str) -> list[dict]:
        if name in self._cache:
            return self._cache[name]
        t = self.catalog.tables.get(name)
        if t is not None and t.as_select:
            rows = self.run_select(t.as_select, sink=name)
        else:
            rows = self._topic_rows(name)
        self._cache[name] = rows
        return rows

    def _topic_rows(self, name: str) -> list[dict]:
        from ..wire.topics import AvroConsumer
        schema = self.schemas.get(name)
        if schema is not None:
            return [r for _, r in
                    AvroConsumer(self.broker, name, schema).poll()]
        topic = self.broker.topics.get(name)
        if topic is None:
            return []
        out = []
        for rec in topic.read_all():
            v = rec.value
            out.append(v if isinstance(v, dict) else {"value": v})
        return out

    def run_table(self, name: str) -> list[dict]:
        """Execute the CTAS behind `name` (and everything it depends on)."""
        self._cache.pop(name, None)
        return self.table_rows(name)

    def run_inserts(self, values_only: bool = False) -> None:
        """Apply catalog INSERT statements (VALUES and INSERT..SELECT).
        values_only=True applies just the VALUES inserts — streaming
        pipelines (sql/stream.py) own the INSERT..SELECT statements."""
        for ins in self.catalog.inserts:
            topic = self.broker.create_topic(ins.table)
            if ins.values:
                cols = [c.name for c in
                        self.catalog.tables[ins.table].columns]
                for vals in ins.values:
                    topic.append(dict(zip(cols, vals)), partition=0)
            elif ins.select and not values_only:
                for row in self.run_select(ins.select, sink=ins.table):
                    topic.append(row, partition=0)
            self._cache.pop(ins.table, None)

    # -- scalar operator forms (walkthrough smoke statements:
    # LAB1-Walkthrough.md:66-92 SELECT ML_PREDICT / AI_TOOL_INVOKE) ------
    def _fn_ml_predict(self, args: list[str], row: _Row):
        model_name = args[0].strip().strip("'")
        model = self.catalog.models.get(model_name)
        text = str(self.ev.eval(args[1], row))
        if model and model.outputs and \
                "ARRAY" in model.outputs[0].type.upper():
            if self.embedder is None:
                raise SqlExecError("no embedder configured")
            return self.embedder.embed(text).tolist()
        if self.llm_batch is None:
            raise SqlExecError("no LLM configured")
        return self.llm_batch([text], [self.max_new_tokens])[0]

    def _fn_tool_invoke(self, args: list[str], row: _Row):
        from ..agents.runner import ai_tool_invoke
        if self.llm_batch is None:
            raise SqlExecError("no LLM configured")
        prompt = str(self.ev.eval(args[1], row))
        tools = self.ev.eval(args[3], row) if len(args) > 3 else {}
        return ai_tool_invoke(
            lambda p, t: self.llm_batch([p], [t])[0],
            self.tool_fn or (lambda n, a: ""), prompt, tools or {})

    def explain(self, table: str) -> list[str]:
        """Human-readable stage plan for a CTAS (what run_table will do)."""
        t = self.catalog.tables[table]
        if not t.as_select:
            return [f"scan topic {table}"]
        info = analyze_select(t.as_select)
        clauses = _split_clauses(t.as_select)
        from_clause, laterals = _extract_laterals(clauses["from"])
        plan = []
        if info.tumble:
            plan.append(
                f"tumble {info.tumble['table']} every "
                f"{info.tumble['window_ms']} ms on {info.tumble['ts_col']}")
            if info.anomaly:
                plan.append(f"anomaly-detect {info.anomaly[0]}")
        else:
            tables = _parse_joins(from_clause)
            plan.append(f"scan {tables[0][0]}")
            for name, alias, cond in tables[1:]:
                plan.append(f"hash-join {name} [{alias}] on {cond}")
            if clauses["group_by"]:
                plan.append(f"group-by {clauses['group_by']}")
        for lat in laterals:
            plan.append(f"lateral {lat['call'].split('(')[0].strip()}"
                        f"{' AS ' + lat['alias'] if lat['alias'] else ''}")
        for key in ("where", "having"):
            if clauses[key]:
                plan.append(f"{key} {clauses[key]}")
        if clauses["order_by"]:
            plan.append(f"order-by {clauses['order_by']}")
        if clauses["limit"]:
            plan.append(f"limit {clauses['limit']}")
        plan.append(f"project -> {table}")
        return plan

    # -- core ---------------------------------------------------------------
    def run_select(self, select_sql: str, sink: str | None = None
                   ) -> list[dict]:
        info = analyze_select(select_sql)
        clauses = _split_clauses(select_sql)
        from_clause, laterals = _extract_laterals(clauses["from"] or "")
        items = _parse_select_items(clauses["select"])

        stage_name = "tumble" if info.tumble else "scan_join"
        with self.tracer.stage(f"{sink or 'select'}:{stage_name}") as sp:
            if clauses["from"] is None:
                rows = [_Row({})]            # scalar SELECT (smoke tests)
            elif info.tumble:
                rows = self._tumble_rows(info, items, select_sql)
            else:
                rows = self._join_rows(from_clause)
                if clauses["group_by"]:      # plain (non-windowed) GROUP BY
                    rows = self._group_rows(rows, clauses["group_by"],
                                            items)
            if sp:
                sp.records_out = len(rows)

        for lat in laterals:
            op = lat["call"].split("(")[0].strip().lower()
            with self.tracer.stage(f"{sink or 'select'}:{op}",
                                   records_in=len(rows)) as sp:
                rows = self._apply_lateral(lat, rows, items, select_sql)
                if sp:
                    sp.records_out = len(rows)

        for cond_key in ("where", "having"):
            cond = clauses[cond_key]
            if cond:
                rows = [r for r in rows if self.ev.pred(cond, r)]

        if clauses["order_by"]:
            rows = self._order(rows, clauses["order_by"])
        if clauses["limit"]:
            rows = rows[: int(clauses["limit"].split()[0])]

        out = [self._project(items, r) for r in rows]
        if sink is not None:
            topic = self.broker.create_topic(sink)
            for row in out:
                topic.append(row, partition=0)
        return out

    def _group_rows(self, rows: list[_Row], group_by: str,
                    items) -> list[_Row]:
        """Plain GROUP BY over a bounded row set: key exprs + the
        COUNT/SUM/AVG select aggregates, one output row per group."""
        key_exprs = [e.strip() for e in P._split_top(group_by)]
        aggs = {}
        for expr, alias in items:
            m = re.match(r"(COUNT|SUM|AVG)\s*\(\s*(.*)\s*\)$", expr.strip(),
                         re.IGNORECASE | re.DOTALL)
            if m:
                aggs[alias] = (m.group(1).upper(), m.group(2).strip())
        groups: dict[tuple, list[_Row]] = {}
        for r in rows:
            k = tuple(self.ev.eval(e, r) for e in key_exprs)
            groups.setdefault(k, []).append(r)
        out = []
        for k, members in groups.items():
            cols = {e.strip("`").split(".")[-1]: v
                    for e, v in zip(key_exprs, k)}
            for alias, (fn, arg) in aggs.items():
                if fn == "COUNT":
                    cols[alias] = len(members)
                else:
                    vals = [float(self.ev.eval(arg, m)) for m in members]
                    cols[alias] = (sum(vals) if fn == "SUM"
                                   else sum(vals) / max(len(vals), 1))
            out.append(_Row({"_g": cols}))
        return out

    def _order(self, rows: list[_Row], order_by: str) -> list[_Row]:
        """Stable multi-key ORDER BY col [ASC|DESC], applied last-first."""
        specs = []
        for item in P._split_top(order_by):
            toks = item.strip().split()
            desc = toks[-1].upper() == "DESC"
            expr = (" ".join(toks[:-1])
                    if toks[-1].upper() in ("ASC", "DESC") else item.strip())
            specs.append((expr, desc))
        for expr, desc in reversed(specs):
            rows = sorted(rows, key=lambda r: self.ev.eval(expr, r),
                          reverse=desc)
        return rows

    def watermark_delay_ms(self, table: str) -> int:
        """Delay from the table's WATERMARK clause (ts - INTERVAL 'n' U);
        5 s when the table declares none (the lab default)."""
        t = self.catalog.tables.get(table)
        if t is not None and t.watermark:
            m = re.search(r"INTERVAL\s+'(\d+)'\s+(\w+)", t.watermark[1],
                          re.IGNORECASE)
            if m:
                from .catalog import _interval_ms
                return _interval_ms(m.group(1), m.group(2))
        return 5000

    # -- FROM stage ---------------------------------------------------------
    def _join_rows(self, from_clause: str) -> list[_Row]:
        tables = _parse_joins(from_clause)
        if not tables:
            raise SqlExecError(f"empty FROM: {from_clause!r}")
        name0, alias0, _ = tables[0]
        rows = [_Row({alias0: r}) for r in self.table_rows(name0)]
        for name, alias, cond in tables[1:]:
            right = self.table_rows(name)
            if cond is None:               # comma join = cross join
                crossed = []
                for r in rows:
                    for rr in right:
                        c = r.child()
                        c.ns[alias] = rr
                        crossed.append(c)
                rows = crossed
                continue
            terms = _split_bool(cond)
            eq_pairs, residual = [], []
            for t in terms:
                m = re.match(r"\s*([`\w.]+)\s*=\s*([`\w.]+)\s*$", t)
                if m:
                    l, r_ = m.group(1).strip("`"), m.group(2).strip("`")
                    if l.startswith(alias + "."):
                        eq_pairs.append((r_, l.split(".", 1)[1]))
                        continue
                    if r_.startswith(alias + "."):
                        eq_pairs.append((l, r_.split(".", 1)[1]))
                        continue
                residual.append(t)
            index: dict[tuple, list[dict]] = {}
            for rr in right:
                key = tuple(rr.get(col) for _, col in eq_pairs)
                index.setdefault(key, []).append(rr)
            joined = []
            for row in rows:
                key = tuple(self.ev.eval(l, row) for l, _ in eq_pairs)
                for rr in index.get(key, ()):
                    cand = row.child()
                    cand.ns[alias] = rr
                    if all(self.ev.pred(t, cand) for t in residual):
                        joined.append(cand)
            rows = joined
        return rows

    # -- TUMBLE + anomaly stage ---------------------------------------------
    def _tumble_rows(self, info, items, select_sql: str) -> list[_Row]:
        from ..runtime.anomaly import AnomalyDetector
        from ..runtime.windows import TumblingWindows

        tum = info.tumble
        src = self.table_rows(tum["table"])
        key_col = None
        mo = re.search(r"OVER\s*\(\s*PARTITION\s+BY\s+([`\w]+)", select_sql,
                       re.IGNORECASE)
        if mo:
            key_col = mo.group(1).strip("`")
        else:
            mg = re.search(r"GROUP\s+BY\s+([`\w]+)", select_sql,
                           re.IGNORECASE)
            if mg:
                key_col = mg.group(1).strip("`")
        if key_col is None:
            raise SqlExecError("TUMBLE without PARTITION BY / GROUP BY key")

        # aggregate select items: COUNT(*) / SUM(expr) / AVG(expr)
        aggs = {}
        for expr, alias in items:
            m = re.match(r"(COUNT|SUM|AVG)\s*\(\s*(.*)\s*\)$", expr.strip(),
                         re.IGNORECASE | re.DOTALL)
            if m:
                fn, arg = m.group(1).upper(), m.group(2).strip()
                aggs[alias] = (fn, arg)

        tw = TumblingWindows(
            tum["window_ms"], key_fn=lambda r: r[key_col],
            ts_fn=lambda r: r[tum["ts_col"]],
            watermark_delay_ms=self.watermark_delay_ms(tum["table"]))
        panes = tw.feed(src) + tw.flush()
        panes.sort(key=lambda p: (p.window_start, str(p.key)))
        rows = []
        for p in panes:
            cols = {key_col: p.key, "window_start": p.window_start,
                    "window_end": p.window_end, "window_time": p.window_time}
            for alias, (fn, arg) in aggs.items():
                if fn == "COUNT":
                    cols[alias] = len(p.rows)
                else:
                    vals = [float(self.ev.eval(
                        arg, _Row({"_": rr}))) for rr in p.rows]
                    cols[alias] = (sum(vals) if fn == "SUM"
                                   else sum(vals) / max(len(vals), 1))
            rows.append(_Row({"_w": cols}))

        if info.anomaly:
            ma = re.search(
                r"ML_DETECT_ANOMALIES\s*\(", select_sql, re.IGNORECASE)
            close = P._find_matching_paren(select_sql, ma.end() - 1)
            val_expr = P._split_top(select_sql[ma.end():close])[0]
            mal = re.search(r"\)\s*AS\s+(\w+)",
                            select_sql[close:], re.IGNORECASE)
            a_alias = mal.group(1) if mal else "anomaly"
            det = AnomalyDetector.from_json_params(info.anomaly[0])
            for row in rows:
                res = det.update(row.ns["_w"][key_col],
                                 float(self.ev.eval(val_expr, row)))
                row.ns[a_alias] = {
                    "forecast_value": res.forecast_value,
                    "upper_bound": res.upper_bound,
                    "lower_bound": res.lower_bound,
                    "is_anomaly": res.is_anomaly,
                }
        return rows

    # -- LATERAL stages (vectorized) ----------------------------------------
    def _apply_lateral(self, lat: dict, rows: list[_Row], items,
                       select_sql: str) -> list[_Row]:
        call = lat["call"]
        m = re.match(r"(\w+)\s*\(", call)
        fn = m.group(1).upper()
        close = P._find_matching_paren(call, m.end() - 1)
        args = P._split_top(call[m.end():close])
        if fn == "ML_PREDICT":
            return self._lateral_ml_predict(args, lat, rows)
        if fn == "VECTOR_SEARCH_AGG":
            return self._lateral_vector_search(args, lat, rows, select_sql)
        if fn == "AI_RUN_AGENT":
            return self._lateral_run_agent(args, lat, rows)
        raise SqlExecError(f"unsupported LATERAL call {fn}")

    def _lateral_ml_predict(self, args, lat, rows):
        model_name = args[0].strip().strip("'")
        model = self.catalog.models.get(model_name)
        is_embed = bool(model and model.outputs and
                        "ARRAY" in model.outputs[0].type.upper())
        alias = lat["alias"] or "ml_predict"
        texts = [str(self.ev.eval(args[1], r)) for r in rows]
        if is_embed:
            if self.embedder is None:
                raise SqlExecError("no embedder configured")
            vecs = self.embedder.embed_batch(texts) if texts else []
            for r, v in zip(rows, vecs):
                r.ns[alias] = {"embedding": v}
                r.last_embedding = v
        else:
            if self.llm_batch is None:
                raise SqlExecError("no LLM configured")
            outs = self.llm_batch(texts, [self.max_new_tokens] * len(texts)) \
                if texts else []
            for r, o in zip(rows, outs):
                r.ns[alias] = {"response": o}
        return rows

    def _lateral_vector_search(self, args, lat, rows, select_sql):
        table = args[0].strip().strip("`")
        k = int(args[3])
        index = self.indexes.get(table)
        if index is None:
            raise SqlExecError(f"no vector index bound for table {table!r}")
        alias = lat["alias"] or "r"
        import numpy as np
        queries = []
        for r in rows:
            try:
                q = self.ev.eval(args[2], r)
            except SqlExecError:
                q = r.last_embedding
            if q is None:
                raise SqlExecError(
                    f"no query embedding for VECTOR_SEARCH_AGG({table})")
            queries.append(np.asarray(q, dtype=np.float32))
        all_hits = (index.search_batch(np.stack(queries), k)
                    if queries else [])
        numbered = re.search(rf"\b{alias}\.(chunk|score|document_id)\d",
                             select_sql)
        if numbered:
            for r, hits in zip(rows, all_hits):
                ns = {}
                for i, h in enumerate(hits, start=1):
                    ns[f"chunk{i}"] = h.chunk
                    ns[f"score{i}"] = h.score
                    ns[f"document_id{i}"] = h.document_id
                    for mk, mv in h.metadata.items():
                        ns[f"{mk}{i}"] = mv
                r.ns[alias] = ns
            return rows
        out = []
        for r, hits in zip(rows, all_hits):      # row per hit (lab2 AS r)
            for h in hits:
                child = r.child()
                child.ns[alias] = {"document_id": h.document_id,
                                   "chunk": h.chunk, "score": h.score,
                                   **h.metadata}
                out.append(child)
        return out

    def _lateral_run_agent(self, args, lat, rows):
        from ..agents.runner import episode
        from ..agents.schedule import run_episodes
        agent_name = args[0].strip().strip("'").strip("`")
        spec = self.catalog.agent_spec(agent_name)
        policy_fn = self.agent_policies.get(agent_name)
        # MAP['debug','true'] anywhere in the call -> per-episode traces
        # (LAB1-Walkthrough.md:253 debug semantics)
        debug = any(
            str((self.ev.eval(a, rows[0]) or {}).get("debug", "")
                if rows else "").lower() == "true"
            for a in args[2:] if a.strip().upper().startswith("MAP["))
        eps = []
        for r in rows:
            prompt = str(self.ev.eval(args[1], r))
            policy = policy_fn(lambda n, _r=r: _r.resolve(n)) \
                if policy_fn else None
            schemas = self.tool_schemas if policy is None else None
            eps.append(episode(spec, prompt, policy=policy, debug=debug,
                               max_new_tokens=self.max_new_tokens,
                               tool_schemas=schemas))
        if self.llm_batch is None:
            raise SqlExecError("no LLM configured for AI_RUN_AGENT")
        results = run_episodes(eps, self.llm_batch,
                               self.tool_fn or (lambda n, a: ""))
        alias = lat["alias"] or "agent_result"
        cols = lat["cols"] or ["status", "response"]
        for r, res in zip(rows, results):
            ns = {"status": res.status, "response": res.response}
            if debug:
                ns["debug_trace"] = res.trace
            r.ns[alias] = {c: ns.get(c) for c in cols} if cols else ns
        return rows

    # -- projection ---------------------------------------------------------
    def _project(self, items, row: _Row) -> dict:
        out = {}
        for expr, alias in items:
            m = re.match(r"(COUNT|SUM|AVG)\s*\(", expr.strip(),
                         re.IGNORECASE)
            if m:                            # window agg: already computed
                try:
                    out[alias] = row.resolve(alias)
                except KeyError:
                    pass
                continue
            if re.match(r"ML_DETECT_ANOMALIES\s*\(", expr.strip(),
                        re.IGNORECASE):
                try:
                    v = row.resolve(alias)   # namespace dict -> flatten
                except KeyError:
                    v = row.ns.get(alias)
                if isinstance(v, dict):
                    out.update(v)
                continue
            if expr.strip() == "*" or expr.strip().endswith(".*"):
                prefix = expr.strip()[:-2]
                for ns_alias, d in row.ns.items():
                    if prefix and ns_alias != prefix:
                        continue
                    for k, val in d.items():
                        out.setdefault(k, val)
                continue
            v = self.ev.eval(expr, row)
            if isinstance(v, dict):
                out.update(v)
            elif hasattr(v, "tolist"):       # numpy arrays/scalars -> JSON
                out[alias] = v.tolist()
            else:
                out[alias] = v
        return out
