"""SQL-subset parser for the streaming-agent DDL surface.

The reference's user-facing API is Flink SQL: CREATE
TABLE / CONNECTION / MODEL / TOOL / AGENT, SET, INSERT, and CTAS over the
operator functions (ML_PREDICT, VECTOR_SEARCH_AGG, ML_DETECT_ANOMALIES,
AI_TOOL_INVOKE, AI_RUN_AGENT).  This module parses that surface — the
statements users run in the lab walkthroughs — into typed DDL objects the
catalog (sql/catalog.py) maps onto the MI355X runtime: topics, models,
the MCP toolsets and the agent episode machine (agents/runner.py).

It is a DDL/config parser, not a general SQL engine: SELECT bodies are
kept verbatim (plus extraction of the operator calls the runtime
understands).  Reference cites: CREATE TOOL/AGENT grammar
LAB1-Walkthrough.md:141-181; CREATE MODEL terraform/core/main.tf:439-563;
CREATE TABLE with watermark lab3 main.tf:301-312.
"""

from __future__ import annotations

import re
from dataclasses import dataclass, field


# ---------------------------------------------------------------------------
# statement objects
# ---------------------------------------------------------------------------

@dataclass
class SetStmt:
    key: str
    value: str


@dataclass
class Column:
    name: str
    type: str


@dataclass
class CreateTable:
    name: str
    columns: list[Column] = field(default_factory=list)
    watermark: tuple[str, str] | None = None   # (ts_col, delay expr)
    primary_key: list[str] = field(default_factory=list)
    options: dict = field(default_factory=dict)
    as_select: str | None = None               # CTAS body, verbatim
    if_not_exists: bool = False


@dataclass
class CreateConnection:
    name: str
    options: dict = field(default_factory=dict)


@dataclass
class CreateModel:
    name: str
    inputs: list[Column] = field(default_factory=list)
    outputs: list[Column] = field(default_factory=list)
    options: dict = field(default_factory=dict)


@dataclass
class CreateTool:
    name: str
    connection: str = ""
    options: dict = field(default_factory=dict)


@dataclass
class CreateAgent:
    name: str
    model: str = ""
    prompt: str = ""
    tools: list[str] = field(default_factory=list)
    options: dict = field(default_factory=dict)


@dataclass
class InsertInto:
    table: str
    select: str | None = None
    values: list[list[str]] = field(default_factory=list)


@dataclass
class DropStmt:
    kind: str      # TABLE | MODEL | CONNECTION | TOOL | AGENT
    name: str
    if_exists: bool = False


# ---------------------------------------------------------------------------
# lexing helpers
# ---------------------------------------------------------------------------

def strip_comments(sql: str) -> str:
    out = []
    i, n = 0, len(sql)
    while i < n:
        ch = sql[i]
        if ch == "'":                       # string literal ('' escapes)
            j = i + 1
            while j < n:
                if sql[j] == "'" and j + 1 < n and sql[j + 1] == "'":
                    j += 2
                    continue
                if sql[j] == "'":
                    break
                j += 1
            out.append(sql[i:j + 1])
            i = j + 1
        elif sql.startswith("--", i):
            while i < n and sql[i] != "\n":
                i += 1
        elif sql.startswith("/*", i):
            j = sql.find("*/", i + 2)
            i = n if j < 0 else j + 2
        else:
            out.append(ch)
            i += 1
    return "".join(out)


def split_statements(sql: str) -> list[str]:
    """Split on ';' outside string literals; drop empties."""
    sql = strip_comments(sql)
    stmts, cur = [], []
    in_str = False
    i, n = 0, len(sql)
    while i < n:
        ch = sql[i]
        if in_str:
            cur.append(ch)
            if ch == "'":
                if i + 1 < n and sql[i + 1] == "'":
                    cur.append("'")
                    i += 1
                else:
                    in_str = False
        elif ch == "'":
            in_str = True
            cur.append(ch)
        elif ch == ";":
            s = "".join(cur).strip()
            if s:
                stmts.append(s)
            cur = []
        else:
            cur.append(ch)
        i += 1
    s = "".join(cur).strip()
    if s:
        stmts.append(s)
    return stmts


def _unquote(s: str) -> str:
    s = s.strip()
    if len(s) >= 2 and s[0] == "'" and s[-1] == "'":
        return s[1:-1].replace("''", "'")
    return s


def _ident(s: str) -> str:
    return s.strip().strip("`")


def _split_top(s: str, sep: str = ",") -> list[str]:
    """Split on sep at paren/angle/string depth 0.

    '<' only opens generic-type depth when immediately followed by an
    identifier character (MAP<STRING, ...>); a comparison like
    ``a >= 90`` or ``a < 12`` must not unbalance the scan."""
    parts, cur = [], []
    depth = 0
    angle = 0
    in_str = False
    i, n = 0, len(s)
    while i < n:
        ch = s[i]
        if in_str:
            cur.append(ch)
            if ch == "'":
                if i + 1 < n and s[i + 1] == "'":
                    cur.append("'")
                    i += 1
                else:
                    in_str = False
        elif ch == "'":
            in_str = True
            cur.append(ch)
        elif ch in "([":
            depth += 1
            cur.append(ch)
        elif ch in ")]":
            depth -= 1
            cur.append(ch)
        elif ch == "<" and i + 1 < n and (s[i + 1].isalnum()
                                          or s[i + 1] in "_<"):
            angle += 1
            cur.append(ch)
        elif ch == ">" and angle > 0:
            angle -= 1
            cur.append(ch)
        elif ch == sep and depth == 0 and angle == 0:
            parts.append("".join(cur))
            cur = []
        else:
            cur.append(ch)
        i += 1
    parts.append("".join(cur))
    return [p.strip() for p in parts if p.strip()]


def _parse_with_options(seg: str) -> dict:
    """'k'='v', 'k2'='v2' -> dict (keys lowercased)."""
    opts = {}
    for item in _split_top(seg):
        if "=" not in item:
            continue
        k, v = item.split("=", 1)
        opts[_unquote(k).lower()] = _unquote(v)
    return opts


def _find_matching_paren(s: str, open_idx: int) -> int:
    depth = 0
    in_str = False
    i = open_idx
    while i < len(s):
        ch = s[i]
        if in_str:
            if ch == "'":
                if i + 1 < len(s) and s[i + 1] == "'":
                    i += 1          # '' escape: consume both quotes
                else:
                    in_str = False
        elif ch == "'":
            in_str = True
        elif ch == "(":
            depth += 1
        elif ch == ")":
            depth -= 1
            if depth == 0:
                return i
        i += 1
    raise ValueError("unbalanced parens")


def _extract_with(stmt: str) -> tuple[str, dict]:
    """Remove a trailing-or-embedded top-level WITH (...) clause; returns
    (stmt_without_with, options)."""
    m = None
    for m2 in re.finditer(r"\bWITH\s*\(", stmt, re.IGNORECASE):
        # top-level check: not inside a string
        prefix = stmt[:m2.start()]
        if prefix.count("'") % 2 == 0 and _depth_at(stmt, m2.start()) == 0:
            m = m2
            break
    if not m:
        return stmt, {}
    open_idx = m.end() - 1
    close = _find_matching_paren(stmt, open_idx)
    opts = _parse_with_options(stmt[open_idx + 1:close])
    return stmt[:m.start()] + stmt[close + 1:], opts


def _depth_at(s: str, idx: int) -> int:
    depth = 0
    in_str = False
    i = 0
    while i < idx:
        ch = s[i]
        if in_str:
            if ch == "'":
                if i + 1 < len(s) and s[i + 1] == "'":
                    i += 1          # '' escape: consume both quotes
                else:
                    in_str = False
        elif ch == "'":
            in_str = True
        elif ch == "(":
            depth += 1
        elif ch == ")":
            depth -= 1
        i += 1
    return depth


# ---------------------------------------------------------------------------
# statement parsers
# ---------------------------------------------------------------------------

_SET_RE = re.compile(r"^SET\s+'([^']+)'\s*=\s*'([^']*)'\s*$",
                     re.IGNORECASE | re.DOTALL)


@dataclass
class ShowStmt:
    kind: str                    # TABLES | MODELS | CONNECTIONS | ...


@dataclass
class DescribeStmt:
    name: str


@dataclass
class ExplainStmt:
    name: str                    # CTAS table to plan


def parse_statement(stmt: str):
    s = stmt.strip()
    up = s.upper()
    if up.startswith("SHOW"):
        m = re.match(r"SHOW\s+(TABLES|MODELS|CONNECTIONS|TOOLS|AGENTS)\s*$",
                     s, re.IGNORECASE)
        if not m:
            raise ValueError(f"bad SHOW: {s!r}")
        return ShowStmt(m.group(1).upper())
    if up.startswith("EXPLAIN"):
        m = re.match(r"EXPLAIN\s+(\S+)\s*$", s, re.IGNORECASE)
        if not m:
            raise ValueError(f"bad EXPLAIN: {s!r}")
        return ExplainStmt(_ident(m.group(1)))
    if up.startswith("DESCRIBE") or up.startswith("DESC "):
        m = re.match(r"DESC(?:RIBE)?\s+(\S+)\s*$", s, re.IGNORECASE)
        if not m:
            raise ValueError(f"bad DESCRIBE: {s!r}")
        return DescribeStmt(_ident(m.group(1)))
    if up.startswith("SET"):
        m = _SET_RE.match(s)
        if not m:
            raise ValueError(f"bad SET: {s!r}")
        return SetStmt(m.group(1), m.group(2))
    if up.startswith("DROP"):
        m = re.match(r"DROP\s+(TABLE|MODEL|CONNECTION|TOOL|AGENT)\s+"
                     r"(IF\s+EXISTS\s+)?(\S+)", s, re.IGNORECASE)
        if not m:
            raise ValueError(f"bad DROP: {s!r}")
        return DropStmt(m.group(1).upper(), _ident(m.group(3)),
                        bool(m.group(2)))
    if up.startswith("INSERT"):
        m = re.match(r"INSERT\s+INTO\s+(\S+)\s+([\s\S]+)$", s, re.IGNORECASE)
        if not m:
            raise ValueError(f"bad INSERT: {s!r}")
        table, rest = _ident(m.group(1)), m.group(2).strip()
        if rest.upper().startswith("VALUES"):
            vals_seg = rest[6:].strip()
            rows = []
            for grp in re.finditer(r"\(", vals_seg):
                pass
            # parse one or more parenthesized tuples
            i = 0
            while i < len(vals_seg):
                if vals_seg[i] == "(":
                    j = _find_matching_paren(vals_seg, i)
                    rows.append([_unquote(v)
                                 for v in _split_top(vals_seg[i + 1:j])])
                    i = j + 1
                else:
                    i += 1
            return InsertInto(table, values=rows)
        return InsertInto(table, select=rest)
    if up.startswith("CREATE CONNECTION"):
        body, opts = _extract_with(s)
        m = re.match(r"CREATE\s+CONNECTION\s+(IF\s+NOT\s+EXISTS\s+)?(\S+)",
                     body, re.IGNORECASE)
        return CreateConnection(_ident(m.group(2)), opts)
    if up.startswith("CREATE MODEL"):
        body, opts = _extract_with(s)
        m = re.match(r"CREATE\s+MODEL\s+(IF\s+NOT\s+EXISTS\s+)?([`\w.-]+)",
                     body, re.IGNORECASE)
        name = _ident(m.group(2))
        inputs, outputs = [], []
        mi = re.search(r"\bINPUT\s*\(", body, re.IGNORECASE)
        if mi:
            j = _find_matching_paren(body, mi.end() - 1)
            inputs = _parse_columns_simple(body[mi.end():j])
        mo = re.search(r"\bOUTPUT\s*\(", body, re.IGNORECASE)
        if mo:
            j = _find_matching_paren(body, mo.end() - 1)
            outputs = _parse_columns_simple(body[mo.end():j])
        return CreateModel(name, inputs, outputs, opts)
    if up.startswith("CREATE TOOL"):
        body, opts = _extract_with(s)
        m = re.match(r"CREATE\s+TOOL\s+(IF\s+NOT\s+EXISTS\s+)?([`\w.-]+)"
                     r"(?:\s+USING\s+CONNECTION\s+([`'\w.-]+))?",
                     body, re.IGNORECASE)
        conn = _ident(_unquote(m.group(3))) if m.group(3) else ""
        return CreateTool(_ident(m.group(2)), conn, opts)
    if up.startswith("CREATE AGENT"):
        return _parse_create_agent(s)
    if up.startswith("CREATE TABLE"):
        return _parse_create_table(s)
    raise ValueError(f"unsupported statement: {s.split()[0:3]}")


def _parse_columns_simple(seg: str) -> list[Column]:
    cols = []
    for item in _split_top(seg):
        parts = item.split(None, 1)
        cols.append(Column(_ident(parts[0]),
                           parts[1].strip() if len(parts) > 1 else ""))
    return cols


def _parse_create_agent(s: str) -> CreateAgent:
    body, opts = _extract_with(s)
    m = re.match(r"CREATE\s+AGENT\s+(IF\s+NOT\s+EXISTS\s+)?([`\w.-]+)",
                 body, re.IGNORECASE)
    agent = CreateAgent(_ident(m.group(2)), options=opts)
    mm = re.search(r"USING\s+MODEL\s+([`'\w.-]+)", body, re.IGNORECASE)
    if mm:
        agent.model = _ident(_unquote(mm.group(1)))
    mp = re.search(r"USING\s+PROMPT\s+(')", body, re.IGNORECASE)
    if mp:
        # scan the string literal (with '' escapes)
        i = mp.start(1)
        j = i + 1
        while j < len(body):
            if body[j] == "'" and j + 1 < len(body) and body[j + 1] == "'":
                j += 2
                continue
            if body[j] == "'":
                break
            j += 1
        agent.prompt = body[i + 1:j].replace("''", "'")
    mt = re.search(r"USING\s+TOOLS\s+([^\n]+?)(?:\s+USING\s|\s*$)", body,
                   re.IGNORECASE | re.DOTALL)
    if mt:
        agent.tools = [_ident(_unquote(t))
                       for t in _split_top(mt.group(1))]
    return agent


def _parse_create_table(s: str) -> CreateTable:
    body, opts = _extract_with(s)
    m = re.match(r"CREATE\s+TABLE\s+(IF\s+NOT\s+EXISTS\s+)?([`\w.-]+)\s*",
                 body, re.IGNORECASE)
    name = _ident(m.group(2))
    t = CreateTable(name, options=opts, if_not_exists=bool(m.group(1)))
    rest = body[m.end():].strip()
    # CTAS?
    mas = re.match(r"(?:\(([\s\S]*)\)\s*)?AS\s+(SELECT[\s\S]+)$", rest,
                   re.IGNORECASE)
    if mas:
        t.as_select = mas.group(2).strip()
        if mas.group(1):
            t.columns = _parse_columns_simple(mas.group(1))
        return t
    if rest.startswith("("):
        j = _find_matching_paren(rest, 0)
        for item in _split_top(rest[1:j]):
            upi = item.upper()
            if upi.startswith("WATERMARK"):
                mw = re.match(r"WATERMARK\s+FOR\s+([`\w]+)\s+AS\s+([\s\S]+)$",
                              item, re.IGNORECASE)
                t.watermark = (_ident(mw.group(1)), mw.group(2).strip())
            elif upi.startswith("PRIMARY KEY"):
                mk = re.search(r"\(([^)]*)\)", item)
                t.primary_key = [_ident(c) for c in mk.group(1).split(",")]
            else:
                parts = item.split(None, 1)
                t.columns.append(Column(
                    _ident(parts[0]),
                    parts[1].strip() if len(parts) > 1 else ""))
    return t


def parse_script(sql: str) -> list:
    return [parse_statement(s) for s in split_statements(sql)]
