-- Lab 3 — agentic fleet management (reference:
-- terraform/lab3-agentic-fleet-management/main.tf:301-312 ride_requests;
-- LAB3-Walkthrough.md:99-471 user statements).

CREATE TABLE ride_requests (
  request_id STRING,
  customer_email STRING,
  pickup_zone STRING,
  dropoff_zone STRING,
  price DOUBLE,
  passenger_count INT,
  request_ts TIMESTAMP_LTZ(3),
  WATERMARK FOR request_ts AS request_ts - INTERVAL '5' SECOND
);

-- lab3 provisions its own MCP connection + model (lab3 main.tf:162-267)
CREATE CONNECTION `remote-mcp-connection` WITH (
  'type' = 'MCP_SERVER',
  'endpoint' = 'stub://local',
  'transport' = 'STREAMABLE_HTTP'
);

CREATE MODEL remote_mcp_model
INPUT (prompt STRING)
OUTPUT (response STRING)
WITH (
  'provider' = 'local',
  'local.model' = 'llama3-8b',
  'mcp.connection' = 'remote-mcp-connection'
);

CREATE TABLE documents_vectordb_lab3 (
  document_id STRING,
  chunk STRING,
  embedding ARRAY<FLOAT>
) WITH (
  'connector' = 'hbm-vector-index',
  'index' = 'vector_index',
  'embedding.dims' = '1536',
  'similarity' = 'cosine',
  'numCandidates' = '500'
);

CREATE TABLE anomalies_per_zone AS
SELECT pickup_zone, window_time, COUNT(*) AS request_count,
  ML_DETECT_ANOMALIES(CAST(request_count AS DOUBLE), window_time,
    JSON_OBJECT('minTrainingSize' VALUE 286, 'maxTrainingSize' VALUE 7000,
                'confidencePercentage' VALUE 99.9, 'enableStl' VALUE FALSE))
    OVER (PARTITION BY pickup_zone ORDER BY window_time
          RANGE UNBOUNDED PRECEDING) AS anomaly
FROM TABLE(TUMBLE(TABLE ride_requests, DESCRIPTOR(request_ts),
                  INTERVAL '5' MINUTE))
GROUP BY pickup_zone, window_start, window_end, window_time
HAVING anomaly.is_anomaly AND request_count > anomaly.upper_bound;

CREATE TABLE anomalies_enriched WITH ('changelog.mode' = 'append') AS
SELECT rad.pickup_zone, rad.window_time, rad.request_count,
       search_results.chunk1, search_results.chunk2, search_results.chunk3,
       ml_predict.response AS anomaly_reason
FROM anomalies_per_zone rad,
LATERAL TABLE(ML_PREDICT('llm_embedding_model', surge_query)),
LATERAL TABLE(VECTOR_SEARCH_AGG(documents_vectordb_lab3,
                                DESCRIPTOR(embedding), rad.embedding, 3))
  AS search_results,
LATERAL TABLE(ML_PREDICT('llm_textgen_model', summarize_prompt))
  AS ml_predict;

CREATE TOOL lab3_remote_mcp
USING CONNECTION `remote-mcp-connection`
WITH (
  'type' = 'mcp',
  'allowed_tools' = 'http_get, http_post',
  'request_timeout' = '30'
);

CREATE AGENT boat_dispatch_agent
USING MODEL remote_mcp_model
USING PROMPT 'You are a fleet dispatch agent for New Orleans water taxis. Fetch the vessel catalog with http_get, choose at most 8 boats whose combined capacity covers the surge, then POST the dispatch JSON with http_post. Respond in exactly three sections: ''Dispatch Summary:'', ''Dispatch JSON:'' and ''API Response:''.'
USING TOOLS lab3_remote_mcp
WITH ('max_iterations' = '10');

CREATE TABLE completed_actions AS
SELECT ae.pickup_zone, ae.window_time,
  REGEXP_EXTRACT(agent_result.response,
    '\*{0,2}Dispatch Summary:?\*{0,2}\s*\n?([\s\S]*?)\n\s*\*{0,2}Dispatch JSON',
    1) AS dispatch_summary,
  REGEXP_EXTRACT(agent_result.response,
    '\*{0,2}Dispatch JSON:?\*{0,2}\s*\n?([\s\S]*?)\n\s*\*{0,2}API Response',
    1) AS dispatch_json,
  REGEXP_EXTRACT(agent_result.response,
    '\*{0,2}API Response:?\*{0,2}\s*\n?([\s\S]+)', 1) AS api_response
FROM anomalies_enriched ae,
LATERAL TABLE(AI_RUN_AGENT('boat_dispatch_agent', ae.anomaly_reason,
                           ae.pickup_zone))
  AS agent_result(status, response);
