-- Lab 4 — public-sector fraud agents (reference:
-- terraform/lab4-pubsec-fraud-agents/main.tf:55-76 claims table,
-- LAB4-Walkthrough.md:124-446 user statements).

SET 'sql.state-ttl' = '14 d';

-- 17 columns like the reference claims table (lab4 main.tf:55-76)
CREATE TABLE claims (
  claim_id STRING,
  applicant_name STRING,
  city STRING,
  is_primary_residence STRING,
  damage_assessed STRING,
  claim_amount STRING,
  has_insurance STRING,
  insurance_amount STRING,
  claim_narrative STRING,
  assessment_date STRING,
  disaster_date STRING,
  previous_claims_count STRING,
  last_claim_date STRING,
  assessment_source STRING,
  shared_account STRING,
  shared_phone STRING,
  claim_timestamp TIMESTAMP(3),
  WATERMARK FOR claim_timestamp AS claim_timestamp - INTERVAL '5' SECOND
);

CREATE TABLE fema_policies_vectordb (
  document_id STRING,
  chunk STRING,
  embedding ARRAY<FLOAT>,
  pages STRING,
  section_reference STRING,
  title STRING,
  fraud_categories ARRAY<STRING>,
  policy_keywords ARRAY<STRING>,
  char_count INT
) WITH (
  'connector' = 'hbm-vector-index',
  'index' = 'vector_index',
  'embedding.dims' = '1536',
  'similarity' = 'cosine'
);

CREATE TABLE claims_anomalies_by_city AS
SELECT city, window_time,
  SUM(CAST(claim_amount AS DOUBLE)) AS total_claim_amount,
  COUNT(*) AS claim_count,
  ML_DETECT_ANOMALIES(total_claim_amount, window_time,
    JSON_OBJECT('minTrainingSize' VALUE 8, 'maxTrainingSize' VALUE 50,
                'confidencePercentage' VALUE 95.0, 'enableStl' VALUE FALSE))
    OVER (PARTITION BY city ORDER BY window_time
          RANGE UNBOUNDED PRECEDING) AS anomaly
FROM TABLE(TUMBLE(TABLE claims, DESCRIPTOR(claim_timestamp),
                  INTERVAL '6' HOUR))
GROUP BY city, window_start, window_end, window_time
HAVING anomaly.is_anomaly AND total_claim_amount > anomaly.upper_bound;

CREATE TABLE claims_to_investigate AS
SELECT c.claim_id, c.applicant_name, c.city, c.claim_amount,
       c.claim_narrative, c.claim_timestamp,
       c.is_primary_residence, c.damage_assessed, c.has_insurance,
       c.insurance_amount, c.shared_account, c.shared_phone,
       c.previous_claims_count
FROM claims c
JOIN claims_anomalies_by_city a
  ON c.city = a.city
 AND c.claim_timestamp BETWEEN a.window_time - INTERVAL '6' HOUR
                           AND a.window_time
WHERE c.claim_narrative <> ''
LIMIT 10;

-- carry the structured claim through: the fraud agent's checklist needs
-- amounts/residence/insurance/shared-identity fields (LAB4:249-310)
CREATE TABLE claims_to_investigate_with_policies AS
SELECT ci.claim_id, ci.claim_narrative, ci.city, ci.claim_amount,
       ci.is_primary_residence, ci.damage_assessed, ci.has_insurance,
       ci.insurance_amount, ci.shared_account, ci.shared_phone,
       search_results.chunk1, search_results.chunk2, search_results.chunk3
FROM claims_to_investigate ci,
LATERAL TABLE(ML_PREDICT('llm_embedding_model', ci.claim_narrative)),
LATERAL TABLE(VECTOR_SEARCH_AGG(fema_policies_vectordb,
                                DESCRIPTOR(embedding), narrative_embedding,
                                3)) AS search_results;

CREATE AGENT claims_fraud_investigation_agent
USING MODEL llm_textgen_model
USING PROMPT 'You are a FEMA claims fraud investigator. Work through the 9-point checklist (identity, address, duplicate claims, damage consistency, amount reasonableness, narrative specificity, policy coverage, documentation, timing). Your verdict must be one of APPROVE, APPROVE_PARTIAL, REQUEST_DOCS, DENY_INELIGIBLE, DENY_FRAUD. Respond in exactly four plain-text sections: ''Verdict:'', ''Issues Found:'', ''Policy Basis:'' and ''Summary:''.'
WITH ('max_iterations' = '10');

CREATE TABLE claims_reviewed AS
SELECT cip.claim_id,
  REGEXP_EXTRACT(agent_result.response,
    '\*{0,2}Verdict:?\*{0,2}\s*\n?([^\n]+)', 1) AS verdict,
  REGEXP_EXTRACT(agent_result.response,
    '\*{0,2}Issues Found:?\*{0,2}\s*\n?([\s\S]*?)\n\s*\*{0,2}Policy Basis',
    1) AS issues_found,
  REGEXP_EXTRACT(agent_result.response,
    '\*{0,2}Policy Basis:?\*{0,2}\s*\n?([\s\S]*?)\n\s*\*{0,2}Summary',
    1) AS policy_basis,
  REGEXP_EXTRACT(agent_result.response,
    '\*{0,2}Summary:?\*{0,2}\s*\n?([\s\S]+)', 1) AS summary
FROM claims_to_investigate_with_policies cip,
LATERAL TABLE(AI_RUN_AGENT('claims_fraud_investigation_agent',
                           investigation_prompt, MAP['debug','true']))
  AS agent_result(status, response);
