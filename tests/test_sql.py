"""SQL-subset surface tests: the CREATE TABLE/MODEL/TOOL/AGENT grammar the
lab walkthroughs use parses into catalog objects that map
onto the runtime (AgentSpec/ToolSet, window/anomaly params)."""

import pytest

from quickstart_streaming_agents_amd.sql import parse as P
from quickstart_streaming_agents_amd.sql.catalog import (Catalog,
                                                         analyze_select,
                                                         ttl_to_ms)

LAB1_DDL = """
-- session config (LAB1-Walkthrough.md:119-120)
SET 'sql.state-ttl' = '1 HOURS';

CREATE TABLE orders (
  order_id STRING,
  customer_id STRING,
  product_id STRING,
  price DOUBLE,
  order_ts TIMESTAMP_LTZ(3)
) WITH ('changelog.mode' = 'append');

CREATE TABLE ride_requests (
  request_id STRING,
  pickup_zone STRING,
  request_ts TIMESTAMP_LTZ(3),
  WATERMARK FOR request_ts AS request_ts - INTERVAL '5' SECOND
);

CREATE CONNECTION `remote-mcp-connection` WITH (
  'type' = 'MCP_SERVER',
  'endpoint' = 'http://127.0.0.1:9/mcp',
  'transport' = 'STREAMABLE_HTTP'
);

CREATE MODEL llm_textgen_model
INPUT (prompt STRING)
OUTPUT (response STRING)
WITH ('provider' = 'local', 'local.model' = 'llama3-8b');

CREATE MODEL remote_mcp_model
INPUT (prompt STRING)
OUTPUT (response STRING)
WITH ('provider' = 'local', 'mcp.connection' = 'remote-mcp-connection');

CREATE TOOL lab1_remote_mcp
USING CONNECTION `remote-mcp-connection`
WITH ('type' = 'mcp',
      'allowed_tools' = 'http_get, send_email',
      'request_timeout' = '30');

CREATE AGENT price_match_agent
USING MODEL remote_mcp_model
USING PROMPT 'You are a price-match agent. It''s a 3-section format.'
USING TOOLS lab1_remote_mcp
WITH ('max_consecutive_failures' = '2', 'MAX_ITERATIONS' = '10');
"""


def test_lab1_ddl_parses_and_maps():
    cat = Catalog()
    cat.execute(LAB1_DDL)
    assert cat.state_ttl_ms() == 3_600_000
    orders = cat.tables["orders"]
    assert [c.name for c in orders.columns] == [
        "order_id", "customer_id", "product_id", "price", "order_ts"]
    assert orders.columns[3].type.upper() == "DOUBLE"
    rr = cat.tables["ride_requests"]
    assert rr.watermark == ("request_ts",
                            "request_ts - INTERVAL '5' SECOND")
    assert cat.connections["remote-mcp-connection"].options["transport"] == \
        "STREAMABLE_HTTP"
    assert cat.models["llm_textgen_model"].inputs[0].name == "prompt"

    spec = cat.agent_spec("price_match_agent")
    assert spec.model == "remote_mcp_model"
    assert spec.max_iterations == 10
    assert spec.max_consecutive_failures == 2
    assert spec.tools.allowed_tools == ("http_get", "send_email")
    assert spec.tools.request_timeout_s == 30.0
    assert "It's a 3-section format." in spec.prompt


def test_ctas_analysis_lab3_anomaly():
    sql = """
    CREATE TABLE anomalies_per_zone AS
    SELECT pickup_zone, window_time, request_count,
      ML_DETECT_ANOMALIES(CAST(request_count AS DOUBLE), window_time,
        JSON_OBJECT('minTrainingSize' VALUE 286, 'maxTrainingSize' VALUE 7000,
                    'confidencePercentage' VALUE 99.9, 'enableStl' VALUE FALSE))
        OVER (PARTITION BY pickup_zone ORDER BY window_time
              RANGE UNBOUNDED PRECEDING) AS a
    FROM TABLE(TUMBLE(TABLE ride_requests, DESCRIPTOR(request_ts),
                      INTERVAL '5' MINUTE));
    """
    cat = Catalog()
    cat.execute(sql)
    info = cat.ctas_info("anomalies_per_zone")
    assert info.tumble["window_ms"] == 300_000
    assert info.tumble["table"] == "ride_requests"
    assert info.anomaly[0]["minTrainingSize"] == 286
    assert info.anomaly[0]["confidencePercentage"] == 99.9
    assert info.anomaly[0]["enableStl"] is False


def test_ctas_analysis_rag_and_agent():
    sql = """
    CREATE TABLE search_results AS
    SELECT qe.query, r.chunk, r.score
    FROM queries_embed qe
    CROSS JOIN LATERAL TABLE(
      VECTOR_SEARCH_AGG(documents_vectordb_lab2, DESCRIPTOR(embedding),
                        qe.embedding, 3)) AS r;

    CREATE TABLE price_match_results AS
    SELECT order_id,
      REGEXP_EXTRACT(agent_result.response,
        '\\*{0,2}Competitor Price:?\\*{0,2}\\s*([^\\n]+)', 1) AS comp_price,
      agent_result.status
    FROM enriched_orders,
    LATERAL TABLE(AI_RUN_AGENT('price_match_agent', user_prompt, order_id))
      AS agent_result(status, response);

    INSERT INTO queries_embed
    SELECT query, embedding FROM queries,
    LATERAL TABLE(ML_PREDICT('llm_embedding_model', query));
    """
    cat = Catalog()
    cat.execute(sql)
    sr = cat.ctas_info("search_results")
    assert sr.vector_search == [{"table": "documents_vectordb_lab2",
                                 "column": "embedding",
                                 "query_expr": "qe.embedding", "k": 3}]
    pm = cat.ctas_info("price_match_results")
    assert pm.run_agent == ["price_match_agent"]
    assert len(pm.regexp_extract) == 1
    assert "Competitor Price" in pm.regexp_extract[0]
    ins = cat.inserts[0]
    assert ins.table == "queries_embed"
    assert analyze_select(ins.select).ml_predict == ["llm_embedding_model"]


def test_insert_values_and_drop():
    cat = Catalog()
    cat.execute("""
    CREATE TABLE queries (query STRING);
    INSERT INTO queries VALUES ('How do I create a Flink table?');
    DROP TABLE queries;
    """)
    assert cat.inserts[0].values == [["How do I create a Flink table?"]]
    assert "queries" not in cat.tables
    with pytest.raises(KeyError):
        cat.execute("DROP AGENT missing_agent;")
    cat.execute("DROP AGENT IF EXISTS missing_agent;")


def test_ttl_units_and_errors():
    assert ttl_to_ms("1 HOURS") == 3_600_000
    assert ttl_to_ms("14 d") == 14 * 86_400_000
    assert ttl_to_ms("30 MINUTES") == 1_800_000
    with pytest.raises(ValueError):
        ttl_to_ms("soon")
    with pytest.raises(KeyError):
        Catalog().execute(
            "CREATE AGENT a USING MODEL nope USING PROMPT 'x';")


def test_statement_splitting_respects_strings():
    stmts = P.split_statements(
        "SET 'a' = 'x;y'; -- comment; with semicolon\n"
        "CREATE TABLE t (c STRING) /* block; comment */;")
    assert len(stmts) == 2
    st = P.parse_statement(stmts[0])
    assert st.value == "x;y"


def test_parser_edge_cases():
    from quickstart_streaming_agents_amd.sql import parse as P
    # WITH before AS (lab3 anomalies_enriched pattern)
    t = P.parse_statement(
        "CREATE TABLE x WITH ('changelog.mode' = 'append') AS "
        "SELECT a FROM b")
    assert t.as_select.startswith("SELECT")
    assert t.options["changelog.mode"] == "append"
    # quoted value containing parens/semicolons/escaped quote
    c = P.parse_statement(
        "CREATE CONNECTION `c.x` WITH ('endpoint' = 'http://h/p?(a;b)', "
        "'note' = 'it''s fine')")
    assert c.name == "c.x"
    assert c.options["endpoint"] == "http://h/p?(a;b)"
    assert c.options["note"] == "it's fine"
    # nested generic types in columns
    t2 = P.parse_statement(
        "CREATE TABLE t (m MAP<STRING, ARRAY<FLOAT>>, "
        "r ROW<a INT, b STRING>, c DECIMAL(10, 2))")
    assert [col.name for col in t2.columns] == ["m", "r", "c"]
    assert t2.columns[0].type == "MAP<STRING, ARRAY<FLOAT>>"
    # agent with multiple tools
    a = P.parse_statement(
        "CREATE AGENT ag USING MODEL m USING PROMPT 'p' "
        "USING TOOLS t1, t2 WITH ('max_iterations' = '3')")
    assert a.tools == ["t1", "t2"]
    # unsupported statement raises
    import pytest
    with pytest.raises(ValueError):
        P.parse_statement("ALTER TABLE x ADD COLUMN y STRING")


def test_show_and_describe():
    from quickstart_streaming_agents_amd.sql.catalog import Catalog
    cat = Catalog()
    cat.execute(LAB1_DDL)
    assert cat.show("TABLES") == ["orders", "ride_requests"]
    assert cat.show("AGENTS") == ["price_match_agent"]
    assert "remote-mcp-connection" in cat.show("CONNECTIONS")
    cols = cat.describe("orders")
    assert cols[0] == ("order_id", "STRING")
    assert ("price", "DOUBLE") in cols
    # parses as statements too (read-only; apply() is a no-op)
    st = P.parse_statement("SHOW TABLES")
    assert st.kind == "TABLES"
    d = P.parse_statement("DESCRIBE `orders`")
    assert d.name == "orders"
    cat.execute("SHOW MODELS; DESCRIBE orders;")   # no-ops, no error


def test_explain_statement_parses():
    st = P.parse_statement("EXPLAIN anomalies_per_zone")
    assert st.name == "anomalies_per_zone"
    cat = Catalog()
    cat.execute(LAB1_DDL + "; EXPLAIN orders;")   # read-only no-op
