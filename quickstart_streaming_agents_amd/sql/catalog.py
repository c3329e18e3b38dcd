"""Catalog: applies parsed DDL and maps it onto the MI355X runtime.

The reference keeps this state inside Confluent Flink's catalog; here it
is an in-process registry.  CREATE TABLE registers a topic schema
(wire/topics.py semantics), CREATE MODEL names an on-GPU model role
(textgen -> models/llama.py | mixtral.py; embedding -> models/encoder.py),
CREATE TOOL/AGENT build the ToolSet/AgentSpec the episode machine runs
(agents/runner.py).  SET holds session config (state TTL etc.,
LAB1-Walkthrough.md:119-120, LAB4:124).

`analyze_select` classifies a CTAS body by the operator calls it contains
(ML_PREDICT / VECTOR_SEARCH_AGG / ML_DETECT_ANOMALIES / AI_RUN_AGENT /
AI_TOOL_INVOKE / TUMBLE / REGEXP_EXTRACT) so lab pipelines can be
assembled from the same SQL users run (drive tests from the
configs users run).
"""

from __future__ import annotations

import re
from dataclasses import dataclass, field

from ..agents.runner import AgentSpec, ToolSet
from . import parse as P


@dataclass
class SelectInfo:
    """Operator calls found in a CTAS / INSERT SELECT body."""
    source_tables: list[str] = field(default_factory=list)
    ml_predict: list[str] = field(default_factory=list)        # model names
    vector_search: list[dict] = field(default_factory=list)    # {table, k}
    anomaly: list[dict] = field(default_factory=list)          # params
    run_agent: list[str] = field(default_factory=list)         # agent names
    tool_invoke: list[str] = field(default_factory=list)       # model names
    tumble: dict | None = None                                 # {ts, interval}
    regexp_extract: list[str] = field(default_factory=list)    # patterns
    joins: int = 0


def _interval_ms(num: str, unit: str) -> int:
    unit = unit.upper().rstrip("S")
    mult = {"SECOND": 1000, "MINUTE": 60_000, "HOUR": 3_600_000,
            "DAY": 86_400_000, "D": 86_400_000, "H": 3_600_000,
            "MIN": 60_000, "SEC": 1000}[unit]
    return int(num) * mult


def ttl_to_ms(v: str) -> int:
    """'1 HOURS' / '14 d' / '30 MINUTES' -> ms (SET 'sql.state-ttl')."""
    m = re.match(r"\s*(\d+)\s*([a-zA-Z]+)\s*$", v)
    if not m:
        raise ValueError(f"bad ttl {v!r}")
    return _interval_ms(m.group(1), m.group(2))


def analyze_select(select: str) -> SelectInfo:
    info = SelectInfo()
    body = select
    for m in re.finditer(r"\bFROM\s+([`\w.-]+)", body, re.IGNORECASE):
        name = m.group(1).strip("`")
        if name.upper() not in ("TABLE",):
            info.source_tables.append(name)
    info.joins = len(re.findall(r"\bJOIN\s+", body, re.IGNORECASE))
    for m in re.finditer(r"\bML_PREDICT\s*\(\s*'([^']+)'", body,
                         re.IGNORECASE):
        info.ml_predict.append(m.group(1))
    for m in re.finditer(
            r"\bVECTOR_SEARCH_AGG\s*\(\s*([`\w.-]+)\s*,\s*DESCRIPTOR\s*\("
            r"\s*([`\w.-]+)\s*\)\s*,\s*([^,]+),\s*(\d+)", body,
            re.IGNORECASE):
        info.vector_search.append({"table": m.group(1).strip("`"),
                                   "column": m.group(2).strip("`"),
                                   "query_expr": m.group(3).strip(),
                                   "k": int(m.group(4))})
    for m in re.finditer(r"\bML_DETECT_ANOMALIES\s*\(", body, re.IGNORECASE):
        j = P._find_matching_paren(body, m.end() - 1)
        args = body[m.end():j]
        params = {}
        for pm in re.finditer(r"'(\w+)'\s+VALUE\s+([\w.]+)", args,
                              re.IGNORECASE):
            v = pm.group(2)
            params[pm.group(1)] = (float(v) if "." in v
                                   else (v.upper() in ("TRUE",)
                                         if v.upper() in ("TRUE", "FALSE")
                                         else int(v)))
        info.anomaly.append(params)
    for m in re.finditer(r"\bAI_RUN_AGENT\s*\(\s*'?([`\w.-]+)'?", body,
                         re.IGNORECASE):
        info.run_agent.append(m.group(1).strip("`"))
    for m in re.finditer(r"\bAI_TOOL_INVOKE\s*\(\s*'([^']+)'", body,
                         re.IGNORECASE):
        info.tool_invoke.append(m.group(1))
    mt = re.search(r"\bTUMBLE\s*\(\s*(?:TABLE\s+)?([`\w.-]+)\s*,\s*"
                   r"DESCRIPTOR\s*\(\s*([`\w.-]+)\s*\)\s*,\s*INTERVAL\s+"
                   r"'(\d+)'\s+(\w+)", body, re.IGNORECASE)
    if mt:
        info.tumble = {"table": mt.group(1).strip("`"),
                       "ts_col": mt.group(2).strip("`"),
                       "window_ms": _interval_ms(mt.group(3), mt.group(4))}
    for m in re.finditer(r"\bREGEXP_EXTRACT\s*\(\s*[^,]+,\s*'((?:[^']|'')*)'",
                         body, re.IGNORECASE):
        info.regexp_extract.append(m.group(1).replace("''", "'"))
    return info


class Catalog:
    def __init__(self):
        self.tables: dict[str, P.CreateTable] = {}
        self.connections: dict[str, P.CreateConnection] = {}
        self.models: dict[str, P.CreateModel] = {}
        self.tools: dict[str, P.CreateTool] = {}
        self.agents: dict[str, P.CreateAgent] = {}
        self.session: dict[str, str] = {}
        self.inserts: list[P.InsertInto] = []

    # ---- execution -------------------------------------------------------
    def execute(self, sql: str) -> list:
        """Parse + apply every statement; returns the parsed objects."""
        parsed = P.parse_script(sql)
        for st in parsed:
            self.apply(st)
        return parsed

    def show(self, kind: str) -> list[str]:
        store = {"TABLES": self.tables, "MODELS": self.models,
                 "CONNECTIONS": self.connections, "TOOLS": self.tools,
                 "AGENTS": self.agents}[kind.upper()]
        return sorted(store)

    def describe(self, name: str) -> list[tuple[str, str]]:
        """(column, type) rows for a table (Flink DESCRIBE shape)."""
        t = self.tables[name]
        out = [(col.name, col.type) for col in t.columns]
        if t.as_select and not out:
            out = [("(CTAS)", t.as_select[:80] + "...")]
        return out

    def apply(self, st) -> None:
        if isinstance(st, (P.ShowStmt, P.DescribeStmt,
                           P.ExplainStmt)):
            return                       # read-only statements
        if isinstance(st, P.SetStmt):
            self.session[st.key] = st.value
        elif isinstance(st, P.CreateTable):
            if st.if_not_exists and st.name in self.tables:
                return
            self.tables[st.name] = st
        elif isinstance(st, P.CreateConnection):
            self.connections[st.name] = st
        elif isinstance(st, P.CreateModel):
            self.models[st.name] = st
        elif isinstance(st, P.CreateTool):
            if st.connection and st.connection not in self.connections:
                raise KeyError(f"unknown connection {st.connection!r}")
            self.tools[st.name] = st
        elif isinstance(st, P.CreateAgent):
            if st.model and st.model not in self.models:
                raise KeyError(f"unknown model {st.model!r}")
            for t in st.tools:
                if t not in self.tools:
                    raise KeyError(f"unknown tool {t!r}")
            self.agents[st.name] = st
        elif isinstance(st, P.InsertInto):
            self.inserts.append(st)
        elif isinstance(st, P.DropStmt):
            store = {"TABLE": self.tables, "MODEL": self.models,
                     "CONNECTION": self.connections, "TOOL": self.tools,
                     "AGENT": self.agents}[st.kind]
            if st.name in store:
                del store[st.name]
            elif not st.if_exists:
                raise KeyError(f"{st.kind} {st.name!r} does not exist")
        else:
            raise TypeError(f"cannot apply {type(st).__name__}")

    # ---- runtime mapping -------------------------------------------------
    def state_ttl_ms(self) -> int | None:
        v = self.session.get("sql.state-ttl")
        return ttl_to_ms(v) if v else None

    def toolset(self, name: str) -> ToolSet:
        t = self.tools[name]
        allowed = tuple(x.strip() for x in
                        t.options.get("allowed_tools", "").split(",")
                        if x.strip())
        return ToolSet(name=t.name, connection=t.connection,
                       allowed_tools=allowed,
                       request_timeout_s=float(
                           t.options.get("request_timeout", 30)))

    def agent_spec(self, name: str) -> AgentSpec:
        a = self.agents[name]
        opts = {k.lower(): v for k, v in a.options.items()}
        tools = self.toolset(a.tools[0]) if a.tools else None
        return AgentSpec(
            name=a.name, model=a.model, prompt=a.prompt, tools=tools,
            max_iterations=int(opts.get("max_iterations", 10)),
            max_consecutive_failures=int(
                opts.get("max_consecutive_failures", 2)),
            options=opts)

    def ctas_info(self, table: str) -> SelectInfo:
        t = self.tables[table]
        assert t.as_select, f"{table} is not a CTAS"
        return analyze_select(t.as_select)
