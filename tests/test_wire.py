"""Avro codec, Confluent wire framing, registry, topic log."""

import io

import pytest

from quickstart_streaming_agents_amd.wire import (
    AvroConsumer, AvroProducer, Broker, Schema, deserialize, peek_schema_id,
    serialize,
)
from quickstart_streaming_agents_amd.wire.avro import _read_long, _write_long
from quickstart_streaming_agents_amd.labs import schemas


def test_zigzag_roundtrip():
    for n in [0, 1, -1, 63, -64, 64, 127, 128, 1 << 20, -(1 << 20),
              1736966400000, -(1 << 40)]:
        out = io.BytesIO()
        _write_long(out, n)
        assert _read_long(io.BytesIO(out.getvalue())) == n


def test_known_zigzag_bytes():
    # Avro spec examples: 0->00, -1->01, 1->02, -2->03, 2->04
    for n, expected in [(0, b"\x00"), (-1, b"\x01"), (1, b"\x02"),
                        (-2, b"\x03"), (2, b"\x04"), (-64, b"\x7f"),
                        (64, b"\x80\x01")]:
        out = io.BytesIO()
        _write_long(out, n)
        assert out.getvalue() == expected


def test_record_roundtrip_orders():
    s = Schema(schemas.ORDERS)
    rec = {"order_id": "ORD-00001", "customer_id": "CUST-001",
           "product_id": "PROD-001", "price": 249.0, "order_ts": 1736966400000}
    payload = serialize(s, 7, rec)
    assert payload[0] == 0
    assert peek_schema_id(payload) == 7
    sid, decoded = deserialize(s, payload)
    assert sid == 7
    assert decoded == rec


def test_union_null_fields():
    s = Schema(schemas.CLAIMS)
    claim = {"claim_id": "CLM-1", "applicant_name": None, "city": "Naples",
             "is_primary_residence": "Yes", "damage_assessed": None,
             "claim_amount": "12000.00", "has_insurance": None,
             "insurance_amount": None, "claim_narrative": "flooded",
             "assessment_date": None, "disaster_date": None,
             "previous_claims_count": None, "last_claim_date": None,
             "assessment_source": None, "shared_account": None,
             "shared_phone": None, "claim_timestamp": 1000}
    _, decoded = deserialize(s, serialize(s, 1, claim))
    assert decoded == claim


def test_array_and_optional_array():
    s = Schema(schemas.DOCUMENTS)
    doc = {"document_id": "DOC-1", "title": "T", "chunk": "c" * 10,
           "pages": None, "section_reference": "S1",
           "fraud_categories": ["duplicate", "inflated"],
           "policy_keywords": [], "char_count": 10}
    _, decoded = deserialize(s, serialize(s, 2, doc))
    assert decoded["fraud_categories"] == ["duplicate", "inflated"]
    assert decoded["policy_keywords"] == []
    assert decoded["char_count"] == 10


def test_float_array_embedding():
    s = Schema({"type": "array", "items": "float"})
    vec = [0.5, -1.25, 3.0]
    out = io.BytesIO()
    s.write(out, vec)
    assert s.read(io.BytesIO(out.getvalue())) == vec


def test_registry_dedup_and_subjects():
    b = Broker()
    p1 = AvroProducer(b, "orders", schemas.ORDERS)
    p2 = AvroProducer(b, "orders", schemas.ORDERS)
    assert p1.value_schema_id == p2.value_schema_id
    assert "orders-value" in b.registry.subjects()


def test_topic_counts_purge_and_consumer():
    b = Broker()
    prod = AvroProducer(b, "orders", schemas.ORDERS)
    for i in range(5):
        prod.produce({"order_id": f"O{i}", "customer_id": "C", "product_id": "P",
                      "price": 1.0, "order_ts": i * 1000},
                     key=f"O{i}", timestamp_ms=i * 1000, partition=0)
    t = b.topic("orders")
    assert t.message_count() == 5
    cons = AvroConsumer(b, "orders", schemas.ORDERS)
    got = cons.poll()
    assert len(got) == 5
    assert got[0][1]["order_id"] == "O0"
    assert cons.poll() == []  # offsets advanced
    t.purge()
    assert t.message_count() == 0


def test_bad_wire_format_rejected():
    s = Schema(schemas.QUERIES)
    with pytest.raises(ValueError):
        deserialize(s, b"\x01\x00\x00\x00\x07junk")


def test_native_codec_byte_identical():
    """C++ AvroCodec (ops/hip/avro_codec.cpp) must be byte-identical to the
    Python codec on every lab schema and round-trip every record."""
    import pytest

    from quickstart_streaming_agents_amd.labs import schemas as S
    from quickstart_streaming_agents_amd.ops import ext, have_ext
    from quickstart_streaming_agents_amd.wire.avro import Schema, serialize
    if not have_ext():
        pytest.skip("extension not built")
    samples = {
        "ORDERS": {"order_id": "O1", "customer_id": "C1", "product_id": "P1",
                   "price": 12.5, "order_ts": 1726000000000},
        "DOCUMENTS": {"document_id": "d1", "chunk": "hello é",
                      "title": None, "pages": "1-2",
                      "section_reference": None,
                      "fraud_categories": ["a", "b"],
                      "policy_keywords": [], "char_count": 7},
    }
    for name, rec in samples.items():
        defn = getattr(S, name)
        codec = ext().AvroCodec(defn)
        py = Schema(defn)
        # fill any missing optional fields with None
        full = {f["name"]: rec.get(f["name"]) for f in defn["fields"]}
        full.update(rec)
        assert bytes(codec.serialize(9, full)) == serialize(py, 9, full)
        sid, back = codec.deserialize(serialize(py, 9, full))
        assert sid == 9
        for k, v in full.items():
            if isinstance(v, float):
                assert abs(back[k] - v) < 1e-9
            else:
                assert back[k] == v


def test_codec_fuzz_python_vs_native():
    """Property test: random records serialize byte-identically in the
    Python and C++ codecs and round-trip through both."""
    import pytest
    from hypothesis import given, settings, strategies as st

    from quickstart_streaming_agents_amd.ops import ext, have_ext
    from quickstart_streaming_agents_amd.wire.avro import (Schema,
                                                           deserialize,
                                                           serialize)
    if not have_ext():
        pytest.skip("extension not built")
    defn = {
        "type": "record", "name": "fuzz", "fields": [
            {"name": "s", "type": "string"},
            {"name": "n", "type": "long"},
            {"name": "d", "type": "double"},
            {"name": "b", "type": "boolean"},
            {"name": "opt", "type": ["null", "string"], "default": None},
            {"name": "arr", "type": {"type": "array", "items": "long"}},
            {"name": "m", "type": {"type": "map", "values": "string"}},
        ]}
    py = Schema(defn)
    codec = ext().AvroCodec(defn)

    @settings(max_examples=200, deadline=None)
    @given(st.text(max_size=80),
           st.integers(min_value=-(2 ** 62), max_value=2 ** 62),
           st.floats(allow_nan=False, allow_infinity=False),
           st.booleans(),
           st.one_of(st.none(), st.text(max_size=20)),
           st.lists(st.integers(min_value=-10 ** 12, max_value=10 ** 12),
                    max_size=8),
           st.dictionaries(st.text(min_size=1, max_size=10),
                           st.text(max_size=10), max_size=5))
    def roundtrip(s, n, d, b, opt, arr, m):
        rec = {"s": s, "n": n, "d": d, "b": b, "opt": opt, "arr": arr,
               "m": m}
        raw_py = serialize(py, 3, rec)
        raw_cc = bytes(codec.serialize(3, rec))
        assert raw_py == raw_cc
        _, back_py = deserialize(py, raw_cc)
        _, back_cc = codec.deserialize(raw_py)
        assert back_py == back_cc == rec

    roundtrip()


def test_registry_backward_compatibility_enforced():
    """Confluent SR default BACKWARD mode: a new subject version must be
    able to read data written with the previous one."""
    import pytest

    from quickstart_streaming_agents_amd.wire.registry import (
        IncompatibleSchemaError, SchemaRegistry, schema_incompatibilities)
    base = {"type": "record", "name": "o", "fields": [
        {"name": "id", "type": "string"},
        {"name": "price", "type": "double"}]}
    reg = SchemaRegistry()
    reg.register("orders-value", base)

    # add field WITH default -> compatible new version
    ok = {"type": "record", "name": "o", "fields": base["fields"] + [
        {"name": "region", "type": "string", "default": "us"}]}
    assert reg.check_compatible("orders-value", ok) == []
    reg.register("orders-value", ok)

    # add field WITHOUT default -> rejected
    bad = {"type": "record", "name": "o", "fields": base["fields"] + [
        {"name": "must", "type": "string"}]}
    assert "without a default" in reg.check_compatible(
        "orders-value", bad)[0]
    with pytest.raises(IncompatibleSchemaError):
        reg.register("orders-value", bad)

    # removing a writer field is fine backward (reader ignores it)
    narrower = {"type": "record", "name": "o", "fields": [
        {"name": "id", "type": "string"}]}
    reg.register("orders-value", narrower)

    # NONE mode disables the check
    reg2 = SchemaRegistry()
    reg2.set_mode("s-value", "NONE")
    reg2.register("s-value", base)
    reg2.register("s-value", bad)

    # promotions + unions + enum symbol removal
    from quickstart_streaming_agents_amd.wire.avro import Schema
    assert schema_incompatibilities(Schema("double"), Schema("int")) == []
    assert schema_incompatibilities(Schema("int"), Schema("double"))
    assert schema_incompatibilities(
        Schema(["null", "string"]), Schema("string")) == []
    assert schema_incompatibilities(
        Schema("string"), Schema(["null", "string"]))
    old_enum = {"type": "enum", "name": "v", "symbols": ["A", "B"]}
    new_enum = {"type": "enum", "name": "v", "symbols": ["A"]}
    assert "symbols removed" in schema_incompatibilities(
        Schema(new_enum), Schema(old_enum))[0]
    assert schema_incompatibilities(Schema(old_enum), Schema(new_enum)) == []


def test_consumer_resolves_writer_schema_by_id():
    """Confluent decode semantics: a consumer built with the NEW (reader)
    schema reads records written with the OLD schema — the wire schema id
    selects the writer schema from the registry and reader-only fields
    take their defaults."""
    from quickstart_streaming_agents_amd.wire import Broker
    from quickstart_streaming_agents_amd.wire.topics import (AvroConsumer,
                                                             AvroProducer)
    v1 = {"type": "record", "name": "o", "fields": [
        {"name": "id", "type": "string"},
        {"name": "price", "type": "double"}]}
    v2 = {"type": "record", "name": "o", "fields": [
        {"name": "id", "type": "string"},
        {"name": "price", "type": "double"},
        {"name": "region", "type": "string", "default": "us"}]}
    broker = Broker()
    AvroProducer(broker, "orders", v1).produce({"id": "a", "price": 1.5})
    AvroProducer(broker, "orders", v2).produce(
        {"id": "b", "price": 2.5, "region": "eu"})

    rows = [v for _, v in AvroConsumer(broker, "orders", v2).poll()]
    assert rows == [{"id": "a", "price": 1.5, "region": "us"},
                    {"id": "b", "price": 2.5, "region": "eu"}]
    # an OLD-schema consumer still reads new records (extra field dropped)
    old_rows = [v for _, v in AvroConsumer(broker, "orders", v1).poll()]
    assert old_rows == [{"id": "a", "price": 1.5},
                        {"id": "b", "price": 2.5}]


def test_keyed_partitioning_is_murmur2():
    """Default keyed partitioning must be Kafka's murmur2 (deterministic
    across processes) — Python's salted hash() would scatter keys
    differently every run."""
    from quickstart_streaming_agents_amd.parallel.stream_shard import \
        partition_for_key
    from quickstart_streaming_agents_amd.wire.topics import Topic
    t = Topic("t", num_partitions=8)
    keys = [f"customer-{i}@example.com" for i in range(50)]
    for k in keys:
        rec = t.append({"k": k}, key=k)
        assert rec.partition == partition_for_key(k, 8)
    # keyless appends pin partition 0 (the labs' watermark determinism)
    assert t.append({"k": None}).partition == 0


def test_keyed_produce_consume_roundtrip():
    """Key schemas round-trip like value schemas (TopicNameStrategy
    <topic>-key subject): the consumer decodes wire-format keys back."""
    from quickstart_streaming_agents_amd.wire import Broker
    from quickstart_streaming_agents_amd.wire.topics import (AvroConsumer,
                                                             AvroProducer)
    vs = {"type": "record", "name": "v", "fields": [
        {"name": "n", "type": "int"}]}
    ks = {"type": "record", "name": "k", "fields": [
        {"name": "id", "type": "string"}]}
    broker = Broker()
    p = AvroProducer(broker, "t", vs, key_schema=ks)
    p.produce({"n": 1}, key={"id": "a"})
    p.produce({"n": 2}, key={"id": "b"})
    assert "t-key" in broker.registry.subjects()
    out = AvroConsumer(broker, "t", vs, key_schema=ks).poll()
    assert [(r.key["id"], v["n"]) for r, v in out] == [("a", 1), ("b", 2)]
    # without a key schema the raw wire bytes come back untouched
    raw = AvroConsumer(broker, "t", vs).poll()
    assert isinstance(raw[0][0].key, (bytes, bytearray))


def test_native_codec_thread_stress():
    """Race-detection smoke for the NATIVE code paths (SURVEY §5 aux row):
    16 threads hammer one shared AvroCodec (serialize+deserialize) and
    per-thread codecs concurrently; every round-trip must be exact.
    (The C++ codec must be stateless per call — this catches shared
    mutable state the way the reference's hand-rolled locks were tested.)"""
    import threading

    from quickstart_streaming_agents_amd.ops import have_ext
    if not have_ext():
        pytest.skip("extension not built")
    from quickstart_streaming_agents_amd.labs.schemas import ORDERS
    from quickstart_streaming_agents_amd.ops import ext
    from quickstart_streaming_agents_amd.wire.avro import Schema
    codec = ext().AvroCodec(Schema(ORDERS).defn)
    rows = [{"order_id": f"o{i}", "customer_id": f"c{i % 7}",
             "product_id": f"p{i % 17}", "price": float(i) + 0.5,
             "order_ts": i * 1000} for i in range(200)]
    errors: list = []

    def worker(tid: int):
        try:
            for rep in range(30):
                for i, r in enumerate(rows):
                    raw = bytes(codec.serialize(5, r))
                    sid, back = codec.deserialize(raw)
                    assert sid == 5 and back == r, (tid, rep, i)
        except Exception as e:   # pragma: no cover
            errors.append(e)

    threads = [threading.Thread(target=worker, args=(t,))
               for t in range(16)]
    for t in threads:
        t.start()
    for t in threads:
        t.join()
    assert not errors, errors
