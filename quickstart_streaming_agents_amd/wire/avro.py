"""Avro binary codec + Confluent wire format.

Implements the subset of Avro the lab topic schemas use (record, string,
double, float, long, int, boolean, null, bytes, unions, array, map, and the
timestamp-millis logical type) plus the Confluent Schema Registry wire
framing: magic byte 0x00 + 4-byte big-endian schema id + Avro binary body.

Behavioral contract from the reference: scripts/publish_lab3_data.py:96-122
(wire-format decode) and testing/helpers/kafka_helper.py:70-87
(Avro-or-JSON tolerant deserialize).  Written from the Avro spec, not ported.
"""

from __future__ import annotations

import io
import json
import struct
from typing import Any, BinaryIO

MAGIC_BYTE = 0

# ---------------------------------------------------------------------------
# varint / zigzag primitives
# ---------------------------------------------------------------------------


def _write_long(out: BinaryIO, n: int) -> None:
    """Zigzag + base-128 varint (Avro long/int encoding)."""
    n = (n << 1) ^ (n >> 63)
    while True:
        b = n & 0x7F
        n >>= 7
        if n:
            out.write(bytes((b | 0x80,)))
        else:
            out.write(bytes((b,)))
            return


def _read_long(buf: BinaryIO) -> int:
    shift = 0
    acc = 0
    while True:
        raw = buf.read(1)
        if not raw:
            raise EOFError("truncated varint")
        b = raw[0]
        acc |= (b & 0x7F) << shift
        if not (b & 0x80):
            break
        shift += 7
    return (acc >> 1) ^ -(acc & 1)


# ---------------------------------------------------------------------------
# Schema
# ---------------------------------------------------------------------------

_PRIMITIVES = {"null", "boolean", "int", "long", "float", "double", "bytes", "string"}


class Schema:
    """A parsed Avro schema (the subset used by the lab topics)."""

    def __init__(self, defn: Any):
        if isinstance(defn, str):
            if defn in _PRIMITIVES:
                self.type = defn
                self.defn: Any = defn
                return
            defn = json.loads(defn)
        self.defn = defn
        if isinstance(defn, list):  # union
            self.type = "union"
            self.branches = [Schema(b) for b in defn]
            return
        t = defn["type"] if isinstance(defn, dict) else defn
        if isinstance(t, (dict, list)):
            # {"type": {...}} nesting
            inner = Schema(t)
            self.__dict__.update(inner.__dict__)
            return
        self.type = t
        if t == "record":
            self.name = defn.get("name", "record")
            self.fields = [(f["name"], Schema(f["type"]), f) for f in defn["fields"]]
        elif t == "array":
            self.items = Schema(defn["items"])
        elif t == "map":
            self.values = Schema(defn["values"])
        elif t == "enum":
            self.symbols = defn["symbols"]
        elif t not in _PRIMITIVES:
            raise ValueError(f"unsupported Avro type: {t!r}")

    # -- canonical-ish form for registry identity --------------------------
    def canonical(self) -> str:
        return json.dumps(self.defn, sort_keys=True, separators=(",", ":"))

    # ------------------------------------------------------------------ encode
    def write(self, out: BinaryIO, value: Any) -> None:
        t = self.type
        if t == "null":
            return
        if t == "boolean":
            out.write(b"\x01" if value else b"\x00")
        elif t in ("int", "long"):
            _write_long(out, int(value))
        elif t == "float":
            out.write(struct.pack("<f", float(value)))
        elif t == "double":
            out.write(struct.pack("<d", float(value)))
        elif t == "string":
            raw = str(value).encode("utf-8")
            _write_long(out, len(raw))
            out.write(raw)
        elif t == "bytes":
            _write_long(out, len(value))
            out.write(bytes(value))
        elif t == "record":
            for name, fs, fdef in self.fields:
                if name in value:
                    fv = value[name]
                elif "default" in fdef:
                    fv = fdef["default"]
                else:
                    raise KeyError(f"missing field {name!r} for record {self.name}")
                fs.write(out, fv)
        elif t == "array":
            seq = list(value)
            if seq:
                _write_long(out, len(seq))
                for item in seq:
                    self.items.write(out, item)
            _write_long(out, 0)
        elif t == "map":
            items = list(value.items())
            if items:
                _write_long(out, len(items))
                for k, v in items:
                    raw = k.encode("utf-8")
                    _write_long(out, len(raw))
                    out.write(raw)
                    self.values.write(out, v)
            _write_long(out, 0)
        elif t == "union":
            idx = self._union_index(value)
            _write_long(out, idx)
            self.branches[idx].write(out, value)
        elif t == "enum":
            _write_long(out, self.symbols.index(value))
        else:
            raise ValueError(f"cannot encode type {t!r}")

    def _union_index(self, value: Any) -> int:
        for i, b in enumerate(self.branches):
            if value is None and b.type == "null":
                return i
            if value is not None and b.type != "null":
                return i
        raise ValueError(f"no union branch for {value!r}")

    # ------------------------------------------------------------------ decode
    def read(self, buf: BinaryIO) -> Any:
        t = self.type
        if t == "null":
            return None
        if t == "boolean":
            return buf.read(1) != b"\x00"
        if t in ("int", "long"):
            return _read_long(buf)
        if t == "float":
            return struct.unpack("<f", buf.read(4))[0]
        if t == "double":
            return struct.unpack("<d", buf.read(8))[0]
        if t == "string":
            n = _read_long(buf)
            return buf.read(n).decode("utf-8")
        if t == "bytes":
            n = _read_long(buf)
            return buf.read(n)
        if t == "record":
            return {name: fs.read(buf) for name, fs, _ in self.fields}
        if t == "array":
            result = []
            while True:
                n = _read_long(buf)
                if n == 0:
                    break
                if n < 0:  # block with byte size prefix
                    n = -n
                    _read_long(buf)
                for _ in range(n):
                    result.append(self.items.read(buf))
            return result
        if t == "map":
            result = {}
            while True:
                n = _read_long(buf)
                if n == 0:
                    break
                if n < 0:
                    n = -n
                    _read_long(buf)
                for _ in range(n):
                    klen = _read_long(buf)
                    k = buf.read(klen).decode("utf-8")
                    result[k] = self.values.read(buf)
            return result
        if t == "union":
            idx = _read_long(buf)
            return self.branches[idx].read(buf)
        if t == "enum":
            return self.symbols[_read_long(buf)]
        raise ValueError(f"cannot decode type {t!r}")


# ---------------------------------------------------------------------------
# Confluent wire framing
# ---------------------------------------------------------------------------


def serialize(schema: Schema, schema_id: int, value: Any) -> bytes:
    """Encode value in Confluent wire format (magic 0x00 + 4B schema id + body)."""
    out = io.BytesIO()
    out.write(struct.pack(">bI", MAGIC_BYTE, schema_id))
    schema.write(out, value)
    return out.getvalue()


def deserialize(schema: Schema, payload: bytes) -> tuple[int, Any]:
    """Decode a Confluent wire-format payload -> (schema_id, value)."""
    if len(payload) < 5 or payload[0] != MAGIC_BYTE:
        raise ValueError("not Confluent Avro wire format")
    schema_id = struct.unpack(">I", payload[1:5])[0]
    buf = io.BytesIO(payload[5:])
    return schema_id, schema.read(buf)


def peek_schema_id(payload: bytes) -> int | None:
    if len(payload) >= 5 and payload[0] == MAGIC_BYTE:
        return struct.unpack(">I", payload[1:5])[0]
    return None


def project_to_reader(reader: Schema, value):
    """Avro schema-resolution projection: reshape a value decoded with the
    WRITER schema into the READER schema's record shape (reader-only
    fields take their declared defaults; writer-only fields are dropped).
    Registry BACKWARD compatibility (registry.py) guarantees the defaults
    exist.  Recurses through records/arrays/maps; other types pass
    through unchanged."""
    t = reader.type
    if t == "record" and isinstance(value, dict):
        out = {}
        for name, fs, fdef in reader.fields:
            if name in value:
                out[name] = project_to_reader(fs, value[name])
            elif "default" in fdef:
                out[name] = fdef["default"]
            else:
                raise KeyError(
                    f"field {name!r} missing and has no default")
        return out
    if t == "array" and isinstance(value, list):
        return [project_to_reader(reader.items, v) for v in value]
    if t == "map" and isinstance(value, dict):
        return {k: project_to_reader(reader.values, v)
                for k, v in value.items()}
    if t == "union":
        for b in reader.branches:
            if (value is None) == (b.type == "null"):
                return project_to_reader(b, value)
    return value
