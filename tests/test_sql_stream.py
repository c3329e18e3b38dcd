"""Streaming (incremental) SQL execution tests (sql/stream.py): chunked
arrival produces the same final results as batch execution, windows close
on watermark advance (not at end), and a snapshot/restore mid-stream
resumes exactly (the reference's continuous-Flink + replay-from-offset
behavior)."""

import json

import pytest

from quickstart_streaming_agents_amd.agents.mcp import StubMcpServer
from quickstart_streaming_agents_amd.labs.deploy import Deployment
from quickstart_streaming_agents_amd.sql.stream import StreamingPipeline


@pytest.fixture(scope="module")
def mcp():
    with StubMcpServer() as srv:
        yield srv


def _chunked_records(dep, topic_name, n_chunks):
    """Drain a populated topic and return its raw records in n chunks."""
    topic = dep.broker.topics[topic_name]
    recs = topic.read_all()
    topic.purge()
    size = max(1, len(recs) // n_chunks)
    return [recs[i:i + size] for i in range(0, len(recs), size)]


def _replay(dep, topic_name, recs):
    t = dep.broker.topics[topic_name]
    for r in recs:
        t.append(r.value, key=r.key, timestamp_ms=r.timestamp_ms,
                 partition=0)


def test_lab3_streaming_matches_batch(mcp):
    # batch reference
    ref = Deployment(labs=(3,), device="cpu")
    ref.datagen(3)
    batch_rows = ref.run_sql(3, mcp_server=mcp)

    # streaming run: same data arriving in 5 chunks
    dep = Deployment(labs=(3,), device="cpu")
    dep.datagen(3)
    chunks = _chunked_records(dep, "ride_requests", 5)
    pipe = StreamingPipeline(dep.sql_executor(3, mcp_server=mcp))
    incr: list = []
    for ch in chunks:
        _replay(dep, "ride_requests", ch)
        out = pipe.advance()
        incr += out["completed_actions"]
    final = pipe.finish()
    incr += final["completed_actions"]

    assert len(incr) == len(batch_rows) >= 1
    assert [r["pickup_zone"] for r in incr] == \
        [r["pickup_zone"] for r in batch_rows]
    assert all(r["pickup_zone"] == "French Quarter" for r in incr)
    # at least the training-period windows closed before the final flush
    # (incremental watermark closure, not a single end-of-input batch)
    assert pipe.by_sink["anomalies_per_zone"].windows.wm.current > 0


def test_lab1_streaming_join_both_arrival_orders(mcp):
    ref = Deployment(labs=(1,), device="cpu")
    ref.datagen(1)
    want = sorted(r["order_id"] for r in ref.run_sql(1, mcp_server=mcp))

    # orders arrive BEFORE the dim tables: the two-sided join must emit
    # when the late dimension side shows up
    dep = Deployment(labs=(1,), device="cpu")
    dep.datagen(1)
    orders = _chunked_records(dep, "orders", 1)[0]
    customers = _chunked_records(dep, "customers", 1)[0]
    products = _chunked_records(dep, "products", 1)[0]
    pipe = StreamingPipeline(dep.sql_executor(1, mcp_server=mcp))
    got: list = []
    _replay(dep, "orders", orders)
    got += pipe.advance()["price_match_results"]
    assert got == []                      # dims not there yet
    _replay(dep, "customers", customers)
    got += pipe.advance()["price_match_results"]
    _replay(dep, "products", products)
    got += pipe.advance()["price_match_results"]
    assert sorted(r["order_id"] for r in got) == want
    assert all(r["agent_status"] == "SUCCESS" for r in got)


def test_lab3_snapshot_restore_resumes_exactly(mcp):
    # uninterrupted reference
    ref = Deployment(labs=(3,), device="cpu")
    ref.datagen(3)
    chunks_ref = _chunked_records(ref, "ride_requests", 4)
    pipe_ref = StreamingPipeline(ref.sql_executor(3, mcp_server=mcp))
    out_ref: list = []
    for ch in chunks_ref:
        _replay(ref, "ride_requests", ch)
        out_ref += pipe_ref.advance()["completed_actions"]
    out_ref += pipe_ref.finish()["completed_actions"]

    # interrupted: advance 2 chunks, snapshot, restore into a NEW pipeline
    dep = Deployment(labs=(3,), device="cpu")
    dep.datagen(3)
    chunks = _chunked_records(dep, "ride_requests", 4)
    pipe1 = StreamingPipeline(dep.sql_executor(3, mcp_server=mcp))
    out: list = []
    for ch in chunks[:2]:
        _replay(dep, "ride_requests", ch)
        out += pipe1.advance()["completed_actions"]
    snap = json.loads(json.dumps(pipe1.snapshot()))   # must be JSON-safe

    pipe2 = StreamingPipeline(dep.sql_executor(3, mcp_server=mcp))
    pipe2.restore(snap)
    for ch in chunks[2:]:
        _replay(dep, "ride_requests", ch)
        out += pipe2.advance()["completed_actions"]
    out += pipe2.finish()["completed_actions"]

    assert [r["pickup_zone"] for r in out] == \
        [r["pickup_zone"] for r in out_ref]
    assert len(out) == len(out_ref) >= 1


def test_lab2_streaming_query_arrives_late(mcp):
    """queries -> queries_embed (INSERT..SELECT) -> search_results ->
    search_results_response reacts to a query published AFTER the
    pipeline started (continuous-statement behavior)."""
    dep = Deployment(labs=(2,), device="cpu")
    dep.datagen(2)
    ex = dep.sql_executor(2)
    ex.run_inserts(values_only=True)       # the terraform sample INSERT
    pipe = StreamingPipeline(ex)
    out1 = pipe.advance()
    n1 = len(out1["search_results_response"])
    assert n1 >= 1                         # the sample query answered
    # nothing new -> nothing emitted
    assert pipe.advance()["search_results_response"] == []
    # late query flows through the whole chain in one advance
    dep.broker.topics["queries"].append(
        {"query": "What is a watermark in Flink?"}, partition=0)
    out2 = pipe.advance()
    assert len(out2["search_results_response"]) == 1
    assert out2["search_results_response"][0]["response"]


def test_durable_checkpoint_resume(mcp, tmp_path):
    """Pipeline state survives a process 'crash' via the torn-write-safe
    CheckpointStore files (runtime/checkpoint.py)."""
    ref = Deployment(labs=(3,), device="cpu")
    ref.datagen(3)
    chunks_ref = _chunked_records(ref, "ride_requests", 3)
    pipe_ref = StreamingPipeline(ref.sql_executor(3, mcp_server=mcp))
    out_ref: list = []
    for ch in chunks_ref:
        _replay(ref, "ride_requests", ch)
        out_ref += pipe_ref.advance()["completed_actions"]
    out_ref += pipe_ref.finish()["completed_actions"]

    dep = Deployment(labs=(3,), device="cpu")
    dep.datagen(3)
    chunks = _chunked_records(dep, "ride_requests", 3)
    pipe1 = StreamingPipeline(dep.sql_executor(3, mcp_server=mcp))
    out: list = []
    _replay(dep, "ride_requests", chunks[0])
    out += pipe1.advance()["completed_actions"]
    cp = pipe1.checkpoint(str(tmp_path), "lab3")
    assert cp == 1
    del pipe1                                    # "crash"

    pipe2 = StreamingPipeline(dep.sql_executor(3, mcp_server=mcp))
    assert pipe2.resume(str(tmp_path), "lab3")
    for ch in chunks[1:]:
        _replay(dep, "ride_requests", ch)
        out += pipe2.advance()["completed_actions"]
    out += pipe2.finish()["completed_actions"]
    assert [r["pickup_zone"] for r in out] == \
        [r["pickup_zone"] for r in out_ref]
    # fresh pipeline with no checkpoint
    assert not StreamingPipeline(
        dep.sql_executor(3, mcp_server=mcp)).resume(str(tmp_path), "other")


def test_run_stream_deployment_wrapper(mcp):
    dep = Deployment(labs=(4,), device="cpu")
    dep.datagen(4)
    rows = dep.run_stream(4)
    assert rows and len(rows) <= 10
    from quickstart_streaming_agents_amd.agents.parse import LAB4_VERDICTS
    assert all(r["verdict"] in LAB4_VERDICTS for r in rows)


def test_run_stream_lab1_and_lab2(mcp):
    dep1 = Deployment(labs=(1,), device="cpu")
    dep1.datagen(1)
    rows1 = dep1.run_stream(1, mcp_server=mcp)
    assert rows1 and all(r["agent_status"] == "SUCCESS" for r in rows1)

    dep2 = Deployment(labs=(2,), device="cpu")
    dep2.datagen(2)
    rows2 = dep2.run_stream(2)
    assert rows2 and all(r["response"] for r in rows2)


def test_ctas_cycle_detected():
    from quickstart_streaming_agents_amd.sql.catalog import Catalog
    from quickstart_streaming_agents_amd.sql.exec import (SqlExecError,
                                                          SqlExecutor)
    from quickstart_streaming_agents_amd.wire import Broker
    cat = Catalog()
    cat.execute("""
    CREATE TABLE a AS SELECT x.v FROM b x;
    CREATE TABLE b AS SELECT y.v FROM a y;
    """)
    with pytest.raises(SqlExecError, match="cycle"):
        StreamingPipeline(SqlExecutor(cat, Broker()))


def test_pipeline_stats_counters(mcp):
    dep = Deployment(labs=(3,), device="cpu")
    dep.datagen(3)
    pipe = StreamingPipeline(dep.sql_executor(3, mcp_server=mcp))
    pipe.advance()
    pipe.finish()
    stats = {s["sink"]: s for s in pipe.stats()}
    apz = stats["anomalies_per_zone"]
    assert apz["emitted"] >= 1
    assert apz["late_dropped"] >= 0 and apz["open_panes"] == 0
    assert stats["completed_actions"]["emitted"] == apz["emitted"]


def test_lab1_dim_resend_emits_each_pair_once(mcp):
    """Regression: the multi-join cascade used to push
    cascaded right-side results back through later stages, double-buffering
    them — re-sending a dimension row then emitted duplicates."""
    dep = Deployment(labs=(1,), device="cpu")
    dep.datagen(1)
    orders = _chunked_records(dep, "orders", 1)[0]
    products = _chunked_records(dep, "products", 1)[0]
    customers = _chunked_records(dep, "customers", 1)[0]
    pipe = StreamingPipeline(dep.sql_executor(1, mcp_server=mcp))
    got: list = []
    # the advisor's arrival order: orders -> products -> customers
    _replay(dep, "orders", orders)
    got += pipe.advance()["price_match_results"]
    _replay(dep, "products", products)
    got += pipe.advance()["price_match_results"]
    _replay(dep, "customers", customers)
    got += pipe.advance()["price_match_results"]
    ids = [r["order_id"] for r in got]
    assert len(ids) == len(set(ids)) >= 1   # each order exactly once

    # re-send ONE product record: each buffered order of that product
    # must pair with the new copy exactly once (streaming-join semantics)
    from quickstart_streaming_agents_amd.labs.schemas import ORDERS, PRODUCTS
    from quickstart_streaming_agents_amd.wire.topics import AvroConsumer
    first = AvroConsumer(dep.broker, "products", PRODUCTS).poll()[0][0]
    decoded = [v for _, v in AvroConsumer(
        dep.broker, "orders", ORDERS).poll()]
    pid_rec = [v for _, v in AvroConsumer(
        dep.broker, "products", PRODUCTS).poll()][0]
    expect = sum(1 for o in decoded if o["product_id"] == pid_rec["product_id"])
    dep.broker.topics["products"].append(
        first.value, key=first.key, timestamp_ms=first.timestamp_ms,
        partition=0)
    new = pipe.advance()["price_match_results"]
    assert len(new) == expect >= 1
    new_ids = [r["order_id"] for r in new]
    assert len(new_ids) == len(set(new_ids))


def test_raw_topic_late_timestamp_append_delivered_once():
    """Regression: the schema-less incremental source
    sliced the timestamp-sorted read_all() by count, so a late-timestamp
    append re-delivered one record and dropped the new one."""
    from quickstart_streaming_agents_amd.sql.catalog import Catalog
    from quickstart_streaming_agents_amd.sql.exec import SqlExecutor
    from quickstart_streaming_agents_amd.wire import Broker
    cat = Catalog()
    cat.execute("CREATE TABLE sink AS SELECT r.id AS id FROM raw r;")
    broker = Broker()
    t = broker.create_topic("raw")
    pipe = StreamingPipeline(SqlExecutor(cat, broker))
    t.append({"id": 1}, timestamp_ms=100, partition=0)
    t.append({"id": 2}, timestamp_ms=200, partition=0)
    out = pipe.advance()["sink"]
    assert [r["id"] for r in out] == [1, 2]
    # late event-time record appended after processing ts=100,200
    t.append({"id": 3}, timestamp_ms=50, partition=0)
    out = pipe.advance()["sink"]
    assert [r["id"] for r in out] == [3]
    assert pipe.advance()["sink"] == []


def test_state_ttl_bounds_join_buffers():
    """`SET 'sql.state-ttl'` evicts join state older than the TTL behind
    the statement's stream time (LAB1-Walkthrough.md:119-120): buffers
    stay bounded on a long stream, and rows inside the TTL still join."""
    from quickstart_streaming_agents_amd.sql.catalog import Catalog
    from quickstart_streaming_agents_amd.sql.exec import SqlExecutor
    from quickstart_streaming_agents_amd.wire import Broker

    def build(ttl_stmt):
        cat = Catalog()
        cat.execute(f"""
        {ttl_stmt}
        CREATE TABLE ev (k STRING, ev_ts TIMESTAMP_LTZ(3));
        CREATE TABLE dim (k STRING, payload STRING, up_ts TIMESTAMP_LTZ(3));
        CREATE TABLE joined AS
          SELECT e.k AS k, d.payload AS payload
          FROM ev e JOIN dim d ON e.k = d.k;
        """)
        broker = Broker()
        broker.create_topic("ev")
        broker.create_topic("dim")
        return broker, StreamingPipeline(SqlExecutor(cat, broker))

    HOUR = 3_600_000
    broker, pipe = build("SET 'sql.state-ttl' = '1 HOURS';")
    # 50 keyed events spaced 10 min apart -> stream time spans >8 h
    for i in range(50):
        broker.topics["ev"].append(
            {"k": f"k{i}", "ev_ts": i * 10 * 60_000}, partition=0)
        pipe.advance()
    stats = {s["sink"]: s for s in pipe.stats()}["joined"]
    # only events within the last hour of stream time remain buffered
    assert stats["join_buffered"] <= 7
    assert stats["join_evicted"] >= 40
    # a dim row arriving now joins live events, not evicted ones
    broker.topics["dim"].append(
        {"k": "k49", "payload": "live", "up_ts": 49 * 10 * 60_000},
        partition=0)
    broker.topics["dim"].append(
        {"k": "k0", "payload": "stale", "up_ts": 49 * 10 * 60_000},
        partition=0)
    out = pipe.advance()["joined"]
    assert [r["k"] for r in out] == ["k49"]

    # without the SET, nothing is evicted (Flink's default: state forever)
    broker2, pipe2 = build("")
    for i in range(50):
        broker2.topics["ev"].append(
            {"k": f"k{i}", "ev_ts": i * 10 * 60_000}, partition=0)
        pipe2.advance()
    stats2 = {s["sink"]: s for s in pipe2.stats()}["joined"]
    assert stats2["join_buffered"] == 50 and stats2["join_evicted"] == 0


def test_lab4_streaming_matches_batch(mcp):
    """Lab4 (6-h TUMBLE + anomaly + interval join + 14-d state TTL +
    agent + 4x REGEXP_EXTRACT) through the incremental executor, chunked
    arrival == batch execution; Naples-only contract holds."""
    ref = Deployment(labs=(4,), device="cpu")
    ref.datagen(4)
    batch_rows = ref.run_sql(4, mcp_server=mcp)

    dep = Deployment(labs=(4,), device="cpu")
    dep.datagen(4)
    chunks = _chunked_records(dep, "claims", 6)
    pipe = StreamingPipeline(dep.sql_executor(4, mcp_server=mcp))
    incr: list = []
    for ch in chunks:
        _replay(dep, "claims", ch)
        incr += pipe.advance()["claims_reviewed"]
    incr += pipe.finish()["claims_reviewed"]

    assert len(incr) == len(batch_rows) >= 1
    assert sorted(r["claim_id"] for r in incr) == \
        sorted(r["claim_id"] for r in batch_rows)
    # content contract: the reviewed claims are the Naples-anomaly ones
    # (claims_reviewed keeps the claim columns the CTAS projects)
    naples = {k for k in incr[0] if "city" in k.lower()}
    if naples:
        col = sorted(naples)[0]
        assert all(r[col] == "Naples" for r in incr)
    vt = {r["verdict"] for r in incr}
    assert vt <= {"APPROVE", "APPROVE_PARTIAL", "REQUEST_DOCS",
                  "DENY_INELIGIBLE", "DENY_FRAUD"}


def test_streaming_self_join_same_table_twice():
    """A table joined twice in one FROM must feed BOTH stages (the
    per-advance fetch is shared, not consumed twice)."""
    from quickstart_streaming_agents_amd.sql.catalog import Catalog
    from quickstart_streaming_agents_amd.sql.exec import SqlExecutor
    from quickstart_streaming_agents_amd.wire import Broker
    cat = Catalog()
    cat.execute("""
    CREATE TABLE pairs AS
      SELECT e.id AS id, a.v AS av, b.v AS bv
      FROM ev e JOIN dim a ON e.ka = a.k
                JOIN dim b ON e.kb = b.k;
    """)
    broker = Broker()
    broker.create_topic("ev")
    broker.create_topic("dim")
    pipe = StreamingPipeline(SqlExecutor(cat, broker))
    broker.topics["dim"].append({"k": "x", "v": 1}, partition=0)
    broker.topics["dim"].append({"k": "y", "v": 2}, partition=0)
    broker.topics["ev"].append({"id": 7, "ka": "x", "kb": "y"},
                               partition=0)
    out = pipe.advance()["pairs"]
    assert out == [{"id": 7, "av": 1, "bv": 2}]
    assert pipe.advance()["pairs"] == []


def test_lab1_join_snapshot_restore_with_ttl_stamps(mcp):
    """Snapshot/restore round-trips the NEW join-buffer format
    (stream-time stamps + eviction counters + per-partition offsets +
    stream_time): restore mid-stream, finish identically."""
    ref = Deployment(labs=(1,), device="cpu")
    ref.datagen(1)
    want = sorted(r["order_id"] for r in ref.run_sql(1, mcp_server=mcp))

    dep = Deployment(labs=(1,), device="cpu")
    dep.datagen(1)
    orders = _chunked_records(dep, "orders", 2)
    customers = _chunked_records(dep, "customers", 1)[0]
    products = _chunked_records(dep, "products", 1)[0]
    pipe1 = StreamingPipeline(dep.sql_executor(1, mcp_server=mcp))
    got: list = []
    _replay(dep, "customers", customers)
    _replay(dep, "orders", orders[0])
    got += pipe1.advance()["price_match_results"]   # no products yet
    snap = json.loads(json.dumps(pipe1.snapshot()))

    pipe2 = StreamingPipeline(dep.sql_executor(1, mcp_server=mcp))
    pipe2.restore(snap)
    q = pipe2.by_sink["enriched_orders"]
    assert sum(st.size() for _, st in q.joins) > 0   # buffers restored
    _replay(dep, "products", products)
    got += pipe2.advance()["price_match_results"]
    for ch in orders[1:]:
        _replay(dep, "orders", ch)
        got += pipe2.advance()["price_match_results"]
    assert sorted(r["order_id"] for r in got) == want
