"""Extract SQL from walkthrough markdown (sql_extractors.py parity).

The reference's E2E tests run exactly the SQL users run by regex-extracting
```sql blocks from the LAB walkthrough markdown (scripts/common/
sql_extractors.py:283-303, honoring a ```sql no-parse opt-out;
testing/e2e/test_lab1.py:43-87).  Same contract here: tests and the CLI
can drive pipelines from documentation files.
"""

from __future__ import annotations

import re

_BLOCK_RE = re.compile(r"```sql([^\n`]*)\n(.*?)```", re.DOTALL | re.IGNORECASE)


def extract_sql_blocks(markdown: str) -> list[str]:
    """All ```sql fenced blocks, skipping ```sql no-parse ones."""
    out = []
    for m in _BLOCK_RE.finditer(markdown):
        info = m.group(1).strip().lower()
        if "no-parse" in info:
            continue
        body = m.group(2).strip()
        if body:
            out.append(body)
    return out


def extract_statements(markdown: str) -> list[str]:
    """Individual SQL statements from every parseable block."""
    from .parse import split_statements
    stmts: list[str] = []
    for block in extract_sql_blocks(markdown):
        stmts.extend(split_statements(block))
    return stmts
