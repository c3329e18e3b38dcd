"""MI355X-native streaming-agent runtime.

A from-scratch, single-node 8xMI355X re-design of the capabilities of
confluentinc/quickstart-streaming-agents: Kafka-wire-format ingest, a
Flink-SQL-subset pipeline surface (CREATE TABLE/MODEL/CONNECTION/TOOL/AGENT,
ML_PREDICT, VECTOR_SEARCH_AGG, ML_DETECT_ANOMALIES, AI_TOOL_INVOKE,
AI_RUN_AGENT), and an on-GPU execution engine: embedding encoder, HBM-resident
vector search, batched anomaly scoring and an agent-LLM serving engine with
paged-attention decode — all as hand-written HIP/CDNA4 kernels plus
PyTorch-ROCm, scaled with RCCL over xGMI.

The reference's observable behavior (topic schemas, operator semantics,
agent-loop caps, determinism contracts) is the compatibility contract; the
implementation is MI355X-first, not a port.
"""

__version__ = "0.1.0"
