-- Lab 1 — price-match agent (reference semantics:
-- terraform/lab1-tool-calling/main.tf:234-326 base tables;
-- LAB1-Walkthrough.md:119-256 user statements).

SET 'sql.state-ttl' = '1 HOURS';

CREATE TABLE orders (
  order_id STRING,
  customer_id STRING,
  product_id STRING,
  price DOUBLE,
  order_ts TIMESTAMP_LTZ(3)
);

CREATE TABLE products (
  product_id STRING,
  product_name STRING,
  price DOUBLE,
  department STRING,
  updated_at TIMESTAMP_LTZ(3)
);

CREATE TABLE customers (
  customer_id STRING,
  customer_email STRING,
  customer_name STRING,
  state STRING,
  updated_at TIMESTAMP_LTZ(3)
);

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

CREATE TABLE enriched_orders AS
SELECT o.order_id, o.customer_id, o.product_id, o.price, o.order_ts,
       c.customer_email, c.customer_name, p.product_name,
       p.price AS list_price
FROM orders o
JOIN customers c ON o.customer_id = c.customer_id
JOIN products p ON o.product_id = p.product_id;

CREATE TOOL lab1_remote_mcp
USING CONNECTION `remote-mcp-connection`
WITH (
  'type' = 'mcp',
  'allowed_tools' = 'http_get, send_email',
  'request_timeout' = '30'
);

CREATE AGENT price_match_agent
USING MODEL remote_mcp_model
USING PROMPT 'You are a price matching assistant. Steps: (1) fetch the competitor page with http_get; (2) find the closest product and extract its price as XX.XX; (3) if the competitor price is lower than our order price, send the price-match email with send_email. Respond in exactly three sections: ''Competitor Price:'', ''Decision:'' (PRICE_MATCH or NO_MATCH) and ''Summary:''.'
USING TOOLS lab1_remote_mcp
WITH ('max_consecutive_failures' = '2', 'MAX_ITERATIONS' = '10');

CREATE TABLE price_match_results AS
SELECT eo.order_id, eo.product_name, eo.customer_email,
  agent_result.status AS agent_status,
  REGEXP_EXTRACT(agent_result.response,
    '\*{0,2}Competitor Price:?\*{0,2}\s*\n?([^\n]+)', 1) AS competitor_price,
  REGEXP_EXTRACT(agent_result.response,
    '\*{0,2}Decision:?\*{0,2}\s*\n?([^\n]+)', 1) AS decision,
  REGEXP_EXTRACT(agent_result.response,
    '\*{0,2}Summary:?\*{0,2}\s*\n?([\s\S]+)', 1) AS summary
FROM enriched_orders eo,
LATERAL TABLE(AI_RUN_AGENT('price_match_agent', user_prompt, eo.order_id,
                           MAP['debug','true']))
  AS agent_result(status, response);
