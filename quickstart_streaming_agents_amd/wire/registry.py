"""In-process Schema Registry shim.

Stands in for Confluent Schema Registry: assigns monotonically increasing
ids to canonical schema texts, resolves subject -> latest schema,
enforces per-subject compatibility on registration (BACKWARD by default,
matching Confluent Cloud's default mode the reference topics run under),
and round-trips through the wire framing in `avro.py`.  Subjects follow
the TopicNameStrategy the reference publishers use (``<topic>-value`` /
``<topic>-key``).
"""

from __future__ import annotations

import threading

from .avro import Schema


class IncompatibleSchemaError(ValueError):
    """New schema cannot read data written with the subject's latest."""


# writer type -> reader types it may be promoted to (Avro spec resolution)
_PROMOTIONS = {
    "int": {"long", "float", "double"},
    "long": {"float", "double"},
    "float": {"double"},
    "string": {"bytes"},
    "bytes": {"string"},
}


def schema_incompatibilities(reader: Schema, writer: Schema,
                             path: str = "$") -> list[str]:
    """Avro schema-resolution check: can `reader` decode data written with
    `writer`?  Returns human-readable problems (empty = compatible).
    Implements the subset of the spec the lab schemas use: records
    (missing reader field needs a default), unions, arrays, maps, enums
    (writer symbols must survive), and numeric/string promotions."""
    rt, wt = reader.type, writer.type
    if wt == "union":
        # every branch the writer may emit must be readable
        out = []
        for b in writer.branches:
            out += schema_incompatibilities(reader, b, path)
        return out
    if rt == "union":
        if any(not schema_incompatibilities(b, writer, path)
               for b in reader.branches):
            return []
        return [f"{path}: no reader union branch accepts writer "
                f"type {wt!r}"]
    if rt != wt:
        if wt in _PROMOTIONS and rt in _PROMOTIONS[wt]:
            return []
        return [f"{path}: reader type {rt!r} cannot read writer type {wt!r}"]
    if rt == "record":
        out = []
        wfields = {name: fs for name, fs, _ in writer.fields}
        for name, fs, fdef in reader.fields:
            if name in wfields:
                out += schema_incompatibilities(fs, wfields[name],
                                                f"{path}.{name}")
            elif "default" not in fdef:
                out.append(f"{path}.{name}: field added without a default")
        return out
    if rt == "array":
        return schema_incompatibilities(reader.items, writer.items,
                                        f"{path}[]")
    if rt == "map":
        return schema_incompatibilities(reader.values, writer.values,
                                        f"{path}{{}}")
    if rt == "enum":
        missing = [s for s in writer.symbols if s not in reader.symbols]
        if missing:
            return [f"{path}: enum symbols removed: {missing}"]
    return []


class SchemaRegistry:
    def __init__(self) -> None:
        self._lock = threading.Lock()
        self._by_id: dict[int, Schema] = {}
        self._id_by_canonical: dict[str, int] = {}
        self._subjects: dict[str, list[int]] = {}
        self._modes: dict[str, str] = {}
        self._next_id = 1

    def set_mode(self, subject: str, mode: str) -> None:
        """Per-subject compatibility mode: BACKWARD (default) or NONE."""
        if mode not in ("BACKWARD", "NONE"):
            raise ValueError(f"unsupported compatibility mode {mode!r}")
        self._modes[subject] = mode

    def mode(self, subject: str) -> str:
        return self._modes.get(subject, "BACKWARD")

    def check_compatible(self, subject: str,
                         schema: Schema | str | dict | list) -> list[str]:
        """Problems preventing `schema` from reading the subject's latest
        version's data ([] = compatible or no prior version)."""
        if not isinstance(schema, Schema):
            schema = Schema(schema)
        with self._lock:
            versions = self._subjects.get(subject)
            if not versions:
                return []
            latest = self._by_id[versions[-1]]
        return schema_incompatibilities(schema, latest)

    def register(self, subject: str, schema: Schema | str | dict | list) -> int:
        if not isinstance(schema, Schema):
            schema = Schema(schema)
        canonical = schema.canonical()
        with self._lock:
            sid = self._id_by_canonical.get(canonical)
            versions = self._subjects.setdefault(subject, [])
            if (sid is None or sid not in versions) and versions and \
                    self._modes.get(subject, "BACKWARD") == "BACKWARD":
                problems = schema_incompatibilities(
                    schema, self._by_id[versions[-1]])
                if problems:
                    raise IncompatibleSchemaError(
                        f"subject {subject!r}: " + "; ".join(problems))
            if sid is None:
                sid = self._next_id
                self._next_id += 1
                self._id_by_canonical[canonical] = sid
                self._by_id[sid] = schema
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
