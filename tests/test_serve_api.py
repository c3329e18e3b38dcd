"""HTTP serving surface tests (serve_api.py): the four capabilities
(K1/K2/K4/K6 per SURVEY.md 2.4) over FastAPI with the injected CPU stub
engine — same wiring the GPU box serves with EngineLLM."""

import pytest

fastapi = pytest.importorskip("fastapi")
from fastapi.testclient import TestClient  # noqa: E402

from quickstart_streaming_agents_amd.serve_api import (build_lab_app,
                                                       create_app)


@pytest.fixture(scope="module")
def client():
    app = build_lab_app(device="cpu", labs=(1, 2))
    with TestClient(app) as c:
        yield c
    app.state.mcp_server.stop()


def test_healthz_and_status(client):
    assert client.get("/healthz").json() == {"status": "ok"}
    st = client.get("/v1/status").json()
    assert st["capabilities"]["completions"] is True
    assert st["capabilities"]["embeddings"] is True
    assert st["capabilities"]["search"] is True
    assert st["capabilities"]["agents"] == ["price_match_agent"]


def test_completions_single_and_batch(client):
    r = client.post("/v1/completions", json={"prompt": "hello"})
    assert r.status_code == 200
    assert len(r.json()["choices"]) == 1
    r2 = client.post("/v1/completions",
                     json={"prompt": ["a", "b", "c"], "max_tokens": 8})
    assert [c["index"] for c in r2.json()["choices"]] == [0, 1, 2]
    assert all(c["text"] for c in r2.json()["choices"])


def test_embeddings_1536_contract(client):
    r = client.post("/v1/embeddings", json={"input": ["x", "y"]})
    body = r.json()
    assert body["dims"] == 1536
    assert len(body["data"]) == 2
    assert len(body["data"][0]["embedding"]) == 1536


def test_search_text_and_vector(client):
    r = client.post("/v1/search",
                    json={"query": "How do I create a Flink table?", "k": 3})
    hits = r.json()["hits"]
    assert len(hits) == 3
    assert hits[0]["score"] >= hits[-1]["score"]
    # vector query round-trip through an embedding
    emb = client.post("/v1/embeddings",
                      json={"input": hits[0]["chunk"]}).json()
    r2 = client.post("/v1/search",
                     json={"query": emb["data"][0]["embedding"], "k": 1})
    assert r2.json()["hits"][0]["document_id"] == hits[0]["document_id"] or \
        r2.json()["hits"][0]["score"] >= hits[0]["score"] - 1e-5
    # dim mismatch rejected
    assert client.post("/v1/search",
                       json={"query": [0.1, 0.2], "k": 1}).status_code == 400


def test_agent_endpoint_runs_episode(client):
    r = client.post("/v1/agents/price_match_agent",
                    json={"prompt": "Check AirPods Pro price",
                          "record_key": "ORD-1"})
    body = r.json()
    assert body["status"] in ("SUCCESS", "FAILED")
    assert body["response"]
    assert body["iterations"] >= 1
    assert client.post("/v1/agents/nope",
                       json={"prompt": "x"}).status_code == 404


def test_validation_errors():
    app = create_app(lambda ps, ts: ["ok"] * len(ps))
    with TestClient(app) as c:
        assert c.post("/v1/completions",
                      json={"prompt": "x", "max_tokens": 0}).status_code == 422
        assert c.post("/v1/embeddings",
                      json={"input": "x"}).status_code == 503
        assert c.post("/v1/search",
                      json={"query": "x"}).status_code == 503


def test_metrics_endpoint(client):
    client.post("/v1/completions", json={"prompt": "count me"})
    body = client.get("/metrics").text
    assert "qsa_requests_total" in body
    assert 'endpoint="completions"' in body
    assert "qsa_request_seconds" in body


def test_status_reports_index_size(client):
    st = client.get("/v1/status").json()
    assert st["index_docs"] > 0            # the lab2 document index
    # CPU stub LLM has no engine stats block
    assert "engine" not in st


def test_sql_endpoint(client):
    st = client.post("/v1/sql", json={
        "statement": "SELECT ML_PREDICT('llm_textgen_model', 'hi') "
                     "AS answer"}).json()
    assert st["ok"] and st["rows"][0]["answer"]
    ddl = client.post("/v1/sql", json={
        "statement": "CREATE TABLE web_t (a STRING); SHOW TABLES;"}).json()
    assert "web_t" in ddl["show"]
    bad = client.post("/v1/sql", json={"statement": "FROB x"})
    assert bad.status_code == 400
