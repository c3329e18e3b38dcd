"""In-process Schema Registry shim.

Stands in for Confluent Schema Registry: assigns monotonically increasing
ids to canonical schema texts, resolves subject -> latest schema, and
round-trips through the wire framing in `avro.py`.  Subjects follow the
TopicNameStrategy the reference publishers use (``<topic>-value`` /
``<topic>-key``).
"""

from __future__ import annotations

import threading
from typing import Any

from .avro import Schema


class SchemaRegistry:
    def __init__(self) -> None:
        self._lock = threading.Lock()
        self._by_id: dict[int, Schema] = {}
        self._id_by_canonical: dict[str, int] = {}
        self._subjects: dict[str, list[int]] = {}
        self._next_id = 1

    def register(self, subject: str, schema: Schema | str | dict | list) -> int:
        if not isinstance(schema, Schema):
            schema = Schema(schema)
        canonical = schema.canonical()
        with self._lock:
            sid = self._id_by_canonical.get(canonical)
            if sid is None:
                sid = self._next_id
                self._next_id += 1
                self._id_by_canonical[canonical] = sid
                self._by_id[sid] = schema
            versions = self._subjects.setdefault(subject, [])
            if sid not in versions:
                versions.append(sid)
            return sid

    def by_id(self, schema_id: int) -> Schema:
        return self._by_id[schema_id]

    def latest(self, subject: str) -> tuple[int, Schema]:
        versions = self._subjects[subject]
        sid = versions[-1]
        return sid, self._by_id[sid]

    def subjects(self) -> list[str]:
        return sorted(self._subjects)


GLOBAL_REGISTRY = SchemaRegistry()
