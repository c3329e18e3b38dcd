"""CLI + catalog-driven deployment tests: the SQL in labs/sql/ is the
config surface users run (drive tests from the same configs);
outputs must reproduce the reference's content invariants."""

import json

import pytest

from quickstart_streaming_agents_amd.agents.mcp import StubMcpServer
from quickstart_streaming_agents_amd.cli import main
from quickstart_streaming_agents_amd.labs.deploy import Deployment


@pytest.fixture(scope="module")
def mcp():
    srv = StubMcpServer().start()
    yield srv
    srv.stop()


def test_deploy_catalog_objects():
    dep = Deployment()
    cat = dep.catalog
    assert {"orders", "products", "customers", "ride_requests",
            "claims", "queries"} <= set(cat.tables)
    assert {"llm_textgen_model", "llm_embedding_model",
            "remote_mcp_model"} <= set(cat.models)
    spec = cat.agent_spec("price_match_agent")
    assert spec.max_iterations == 10
    assert spec.tools.allowed_tools == ("http_get", "send_email")
    spec3 = cat.agent_spec("boat_dispatch_agent")
    assert spec3.tools.allowed_tools == ("http_get", "http_post")
    spec4 = cat.agent_spec("claims_fraud_investigation_agent")
    assert spec4.tools is None  # pure-reasoning agent (LAB4:330-384)
    # anomaly params parsed from the CTAS
    p3 = cat.ctas_info("anomalies_per_zone").anomaly[0]
    assert p3 == {"minTrainingSize": 286, "maxTrainingSize": 7000,
                  "confidencePercentage": 99.9, "enableStl": False}
    assert cat.ctas_info("anomalies_per_zone").tumble["window_ms"] == 300_000
    p4 = cat.ctas_info("claims_anomalies_by_city").anomaly[0]
    assert p4["minTrainingSize"] == 8 and p4["maxTrainingSize"] == 50


def test_lab1_sql_driven_run(mcp):
    dep = Deployment(labs=(1,))
    dep.datagen(1)
    rows = dep.run(1, mcp_server=mcp)
    assert len(rows) == 10
    assert all(r["agent_status"] == "SUCCESS" for r in rows)
    assert any(r["decision"] == "PRICE_MATCH" for r in rows)


def test_lab3_sql_driven_french_quarter_only(mcp):
    dep = Deployment(labs=(3,))
    dep.datagen(3)
    rows = dep.run(3, mcp_server=mcp)
    assert 1 <= len(rows) <= 2                       # test_lab3.py:248-257
    assert all(r["pickup_zone"] == "French Quarter" for r in rows)
    boats = json.loads(rows[0]["dispatch_json"])["boats"]
    assert len(boats) <= 8


def test_lab4_sql_driven_naples_verdicts(mcp):
    dep = Deployment(labs=(4,))
    dep.datagen(4)
    rows = dep.run(4)
    assert len(rows) == 10                           # LIMIT 10
    assert all(r["city"] == "Naples" for r in rows)  # test_lab4.py:265-274
    allowed = {"APPROVE", "APPROVE_PARTIAL", "REQUEST_DOCS",
               "DENY_INELIGIBLE", "DENY_FRAUD"}
    assert all(r["verdict"] in allowed for r in rows)


def test_cli_commands(tmp_path, capsys):
    assert main(["deploy", "--dir", str(tmp_path)]) == 0
    assert (tmp_path / "DEPLOYED_RESOURCES.md").exists()
    assert (tmp_path / "LAB3_SQL_COMMANDS.md").exists()
    assert main(["validate"]) == 0
    assert main(["datagen", "--lab", "1"]) == 0
    out = capsys.readouterr().out
    assert "orders: " in out
    assert main(["destroy", "--dir", str(tmp_path)]) == 0
    assert not (tmp_path / "DEPLOYED_RESOURCES.md").exists()


def test_deployment_destroy():
    dep = Deployment(labs=(1,))
    dep.datagen(1)
    dep.destroy()
    assert not dep.catalog.tables and not dep.broker.topics


def test_walkthrough_sql_roundtrip(tmp_path):
    """The generated LAB*_SQL_COMMANDS.md artifacts re-parse into the same
    catalog objects (reference pattern: tests run the SQL extracted from
    the walkthrough markdown — testing/README.md:208-211)."""
    from quickstart_streaming_agents_amd.sql.catalog import Catalog
    from quickstart_streaming_agents_amd.sql.extract import extract_statements
    dep = Deployment(labs=(1, 3))
    dep.write_summaries(str(tmp_path))
    cat = Catalog()
    for lab in (1, 3):
        md = (tmp_path / f"LAB{lab}_SQL_COMMANDS.md").read_text()
        stmts = extract_statements(md)
        assert stmts, f"no SQL extracted for lab{lab}"
        for s in stmts:
            from quickstart_streaming_agents_amd.sql import parse as P
            cat.apply(P.parse_statement(s))
    spec = cat.agent_spec("price_match_agent")
    assert spec.tools.allowed_tools == ("http_get", "send_email")
    spec3 = cat.agent_spec("boat_dispatch_agent")
    assert spec3.tools.allowed_tools == ("http_get", "http_post")
    assert cat.ctas_info("anomalies_per_zone").anomaly[0][
        "minTrainingSize"] == 286


def test_model_resolves_from_sql():
    """CREATE MODEL WITH('local.model'=...) drives engine selection when
    no CLI override is given."""
    dep = Deployment(labs=(1,), model=None)
    assert dep.resolved_model() == "llama3-8b"
    dep2 = Deployment(labs=(1,), model="tiny")
    assert dep2.resolved_model() == "tiny"
