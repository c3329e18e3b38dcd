-- Core models + connection surface (reference: terraform/core/main.tf
-- 278-563 creates per-cloud LLM/embedding connections and the two models).
-- Here both models are local MI355X engines: the textgen model is the
-- paged-attention decode engine (models/llama.py or models/mixtral.py),
-- the embedding model the on-GPU encoder (models/encoder.py, 1536-d).

CREATE CONNECTION `local-llm-connection` WITH (
  'type' = 'LOCAL_ENGINE',
  'engine' = 'paged-decode',
  'device' = 'cuda'
);

CREATE MODEL llm_textgen_model
INPUT (prompt STRING)
OUTPUT (response STRING)
WITH (
  'provider' = 'local',
  'local.model' = 'llama3-8b',
  'local.dtype' = 'bf16',
  'local.max_tokens' = '50000'
);

CREATE MODEL llm_embedding_model
INPUT (text STRING)
OUTPUT (embedding ARRAY<FLOAT>)
WITH (
  'provider' = 'local',
  'local.model' = 'bge-small',
  'local.dims' = '1536'
);
