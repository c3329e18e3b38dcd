"""Capture/replay parity: purge-then-publish, chronological sort, and the
window-aligned timestamp rebase that makes the lab3 spike window close
(publish_lab3_data.py:143-170 semantics)."""

from quickstart_streaming_agents_amd.labs import capture, datagen, pipelines, schemas
from quickstart_streaming_agents_amd.wire import Broker

MIN5 = 5 * 60 * 1000


def test_capture_then_replay_roundtrip(tmp_path):
    b = Broker()
    datagen.publish_lab3(b, seed=42)
    n_orig = b.topic("ride_requests").message_count()
    p = tmp_path / "rides.jsonl"
    n_cap = capture.capture_topic(b, "ride_requests", schemas.RIDE_REQUESTS,
                                  str(p))
    assert n_cap == n_orig > 20_000

    b2 = Broker()
    now_ms = 1_800_000_000_000
    n_rep = capture.replay_file(b2, str(p), "ride_requests",
                                schemas.RIDE_REQUESTS, "request_ts",
                                window_ms=MIN5, now_ms=now_ms)
    assert n_rep == n_orig
    # chronological order per partition
    recs = b2.topic("ride_requests").read_all()
    ts = [r.timestamp_ms for r in recs]
    assert ts == sorted(ts)
    # rebased end: 10 s past an aligned 5-min boundary, in the "past"
    assert ts[-1] % MIN5 == 10_000
    assert ts[-1] <= now_ms


def test_replayed_stream_keeps_anomaly_contract(tmp_path):
    b = Broker()
    datagen.publish_lab3(b, seed=42)
    p = tmp_path / "rides.jsonl"
    capture.capture_topic(b, "ride_requests", schemas.RIDE_REQUESTS, str(p))
    b2 = Broker()
    capture.replay_file(b2, str(p), "ride_requests", schemas.RIDE_REQUESTS,
                        "request_ts", window_ms=MIN5,
                        now_ms=1_800_000_000_000)
    rows = pipelines.lab3_anomalies(b2)
    assert 1 <= len(rows) <= 2
    assert all(r["pickup_zone"] == "French Quarter" for r in rows)


def test_replay_purges_first(tmp_path):
    b = Broker()
    datagen.publish_lab3(b, seed=42)
    p = tmp_path / "r.jsonl"
    capture.capture_topic(b, "ride_requests", schemas.RIDE_REQUESTS, str(p))
    n = b.topic("ride_requests").message_count()
    # replay onto the SAME broker: purge-then-publish keeps the count
    capture.replay_file(b, str(p), "ride_requests", schemas.RIDE_REQUESTS,
                        "request_ts", window_ms=MIN5,
                        now_ms=1_800_000_000_000)
    assert b.topic("ride_requests").message_count() == n


def test_publish_docs_and_sql_extract(tmp_path):
    """publish_docs.py + sql_extractors.py parity: frontmatter chunking to
    the documents topic; ```sql extraction with no-parse opt-out."""
    from quickstart_streaming_agents_amd.labs.docs import (doc_records,
                                                           publish_docs)
    from quickstart_streaming_agents_amd.sql.extract import (
        extract_sql_blocks, extract_statements)
    md = tmp_path / "guide.md"
    md.write_text(
        "---\n"
        "title: Flink Windows\n"
        "pages: 3-5\n"
        "fraud_categories: [dup, identity]\n"
        "---\n"
        "# Windows\n" + ("tumbling windows close on watermark. " * 30) +
        "\n# Joins\nstate ttl evicts rows.\n")
    recs = doc_records(str(md), max_chars=400)
    assert len(recs) >= 2
    assert recs[0]["title"] == "Flink Windows"
    assert recs[0]["fraud_categories"] == ["dup", "identity"]
    assert all(r["char_count"] == len(r["chunk"]) for r in recs)

    b = Broker()
    n = publish_docs(b, [str(md)], max_chars=400)
    assert b.topic("documents").message_count() == n == len(recs)
    # index ingests the published docs
    from quickstart_streaming_agents_amd.labs.pipelines import lab2_build_index
    from quickstart_streaming_agents_amd.vector.index import HashingEmbedder
    idx = lab2_build_index(b, HashingEmbedder())
    hits = idx.search(HashingEmbedder().embed("tumbling windows watermark"), 2)
    assert hits and "window" in hits[0].chunk

    walkthrough = (
        "Intro\n```sql\nCREATE TABLE t1 (c STRING);\nSET 'a' = 'b';\n```\n"
        "skip this:\n```sql no-parse\nDROP TABLE nope;\n```\n"
        "```SQL\nCREATE TABLE t2 (d STRING);\n```\n")
    blocks = extract_sql_blocks(walkthrough)
    assert len(blocks) == 2
    stmts = extract_statements(walkthrough)
    assert len(stmts) == 3
    assert "DROP" not in " ".join(stmts)


def test_cli_capture_subcommand(tmp_path):
    from quickstart_streaming_agents_amd.cli import main
    out = tmp_path / "rides.jsonl"
    rc = main(["capture", "--lab", "3", "--out", str(out)])
    assert rc == 0
    lines = out.read_text().strip().splitlines()
    assert len(lines) >= 28_000          # the lab3 volume contract
