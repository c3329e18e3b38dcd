-- Lab 2 — RAG pipeline (reference: terraform/lab2-vector-search/main.tf
-- 108-331; fully terraform-managed topology
-- queries -> queries_embed -> search_results -> search_results_response).

CREATE TABLE queries (query STRING);

CREATE TABLE queries_embed (query STRING, embedding ARRAY<FLOAT>);

-- external vector table: HBM-resident index (the reference points this at
-- MongoDB Atlas $vectorSearch, numCandidates=500, index vector_index;
-- here rows live in GPU HBM and search is the cosine top-k HIP kernel)
CREATE TABLE documents_vectordb_lab2 (
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

INSERT INTO queries VALUES ('How do I create a Flink table?');

INSERT INTO queries_embed
SELECT query, embedding
FROM queries, LATERAL TABLE(ML_PREDICT('llm_embedding_model', query));

-- unpack the top-3 hits into numbered columns (lab2 main.tf:292 shape)
CREATE TABLE search_results AS
SELECT qe.query,
       r.chunk1 AS chunk_1, r.score1 AS score_1,
       r.chunk2 AS chunk_2, r.score2 AS score_2,
       r.chunk3 AS chunk_3, r.score3 AS score_3
FROM queries_embed qe
CROSS JOIN LATERAL TABLE(
  VECTOR_SEARCH_AGG(documents_vectordb_lab2, DESCRIPTOR(embedding),
                    qe.embedding, 3)) AS r;

CREATE TABLE search_results_response AS
SELECT sr.query, ml_predict.response
FROM search_results sr,
LATERAL TABLE(ML_PREDICT('llm_textgen_model', rag_prompt)) AS ml_predict;
