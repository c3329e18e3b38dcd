"""HTTP serving surface for the engine (production-serving face of the
framework).

The reference exposes its capabilities only through Flink SQL statements
against managed endpoints; a local MI355X deployment needs
a network-facing entry point for the same four capabilities, so this
module serves them over HTTP (FastAPI):

    POST /v1/completions   -> K4 ML_PREDICT('llm_textgen_model', ...)
    POST /v1/embeddings    -> K1 ML_PREDICT('llm_embedding_model', ...)
    POST /v1/search        -> K2 VECTOR_SEARCH_AGG(..., k)
    POST /v1/agents/{name} -> K6 AI_RUN_AGENT(agent, prompt[, key])
    GET  /healthz, /v1/status

Everything is dependency-injected: `create_app` takes the same
`llm_batch` callable the lab pipelines use (labs/pipelines.py StubLLM on
CPU, models/serve.py EngineLLM on a GPU), a HashingEmbedder-compatible
embedder, a VectorIndex, and a dict of agent episode callables — so the
app is unit-testable on CPU and serves the hipGraph engine unchanged on
an MI355X.
"""

from __future__ import annotations

import time
from typing import Any, Callable, Optional, Union

from pydantic import BaseModel, Field


class CompletionRequest(BaseModel):
    prompt: Union[str, list[str]]
    max_tokens: int = Field(default=64, ge=1, le=4096)


class EmbeddingRequest(BaseModel):
    input: Union[str, list[str]]


class SearchRequest(BaseModel):
    query: Union[str, list[float]]
    k: int = Field(default=3, ge=1, le=100)


class AgentRequest(BaseModel):
    prompt: str
    record_key: Optional[str] = None


class SqlRequest(BaseModel):
    statement: str
    max_rows: int = Field(default=20, ge=1, le=1000)


def create_app(llm: Callable[[list[str], list[int]], list[str]],
               embedder=None,
               index=None,
               agents: dict[str, Callable[..., Any]] | None = None,
               sql_executor=None):
    """Build the FastAPI app around injected engine components.

    `agents` maps agent name -> callable(prompt, record_key) returning an
    object with .status/.response/.iterations/.tool_calls (the
    agents/runner.py episode result shape).
    """
    from fastapi import FastAPI, HTTPException

    app = FastAPI(title="quickstart-streaming-agents-amd", version="0.1")
    app.state.started = time.time()
    app.state.requests = 0
    agents = agents or {}

    @app.get("/healthz")
    def healthz():
        return {"status": "ok"}

    # per-app Prometheus registry (a fresh one per create_app so tests
    # can build many apps without duplicate-timeseries errors)
    from prometheus_client import (CONTENT_TYPE_LATEST, CollectorRegistry,
                                   Counter, Histogram, generate_latest)
    registry = CollectorRegistry()
    req_count = Counter("qsa_requests_total", "requests per endpoint",
                        ["endpoint"], registry=registry)
    req_latency = Histogram("qsa_request_seconds", "request latency",
                            ["endpoint"], registry=registry)
    app.state.prom_registry = registry

    def _observe(endpoint: str, t0: float) -> None:
        req_count.labels(endpoint).inc()
        req_latency.labels(endpoint).observe(time.time() - t0)

    @app.get("/metrics")
    def metrics():
        """Prometheus exposition (request counts + latency histograms —
        the observability layer the reference delegates to Confluent's
        managed UI)."""
        from fastapi import Response
        return Response(generate_latest(registry),
                        media_type=CONTENT_TYPE_LATEST)

    @app.get("/v1/status")
    def status():
        out = {
            "uptime_s": round(time.time() - app.state.started, 3),
            "requests": app.state.requests,
            "capabilities": {
                "completions": True,
                "embeddings": embedder is not None,
                "search": index is not None and embedder is not None,
                "agents": sorted(agents),
            },
        }
        engine = getattr(llm, "engine", None)
        if engine is not None and hasattr(engine, "stats"):
            st = engine.stats
            out["engine"] = {
                "prefill_tokens": st.prefill_tokens,
                "cached_prefix_tokens": st.cached_prefix_tokens,
                "decode_tokens": st.decode_tokens,
                "decode_steps": st.decode_steps,
                "prefill_batches": st.prefill_batches,
                "jump_forward_tokens": st.jump_forward_tokens,
            }
        if index is not None:
            out["index_docs"] = len(index)
        out["capabilities"]["sql"] = sql_executor is not None
        return out

    @app.post("/v1/sql")
    def run_sql(req: SqlRequest):
        """Execute one statement (docs/SQL.md grammar) against the
        deployment's catalog/topics: DDL applies, CTAS materializes and
        returns rows, scalar SELECTs evaluate."""
        app.state.requests += 1
        if sql_executor is None:
            raise HTTPException(503, "no SQL executor configured")
        from .sql import parse as P
        from .sql.exec import SqlExecError
        t0 = time.time()
        cat = sql_executor.catalog
        try:
            result: dict = {"ok": True}
            if req.statement.strip().rstrip(";").upper().startswith(
                    "SELECT"):
                rows = sql_executor.run_select(
                    req.statement.strip().rstrip(";"))
                result["row_count"] = len(rows)
                result["rows"] = rows[: req.max_rows]
            else:
                for st in P.parse_script(req.statement):
                    cat.apply(st)
                    if isinstance(st, P.ShowStmt):
                        result["show"] = cat.show(st.kind)
                    elif isinstance(st, P.DescribeStmt):
                        result["describe"] = cat.describe(st.name)
                    elif isinstance(st, P.InsertInto):
                        sql_executor.run_inserts()
                        cat.inserts.clear()
                    elif isinstance(st, P.CreateTable) and st.as_select:
                        rows = sql_executor.run_table(st.name)
                        result["table"] = st.name
                        result["row_count"] = len(rows)
                        result["rows"] = rows[: req.max_rows]
        except (ValueError, SqlExecError, KeyError) as e:
            raise HTTPException(400, f"{type(e).__name__}: {e}")
        _observe("sql", t0)
        return result

    @app.post("/v1/completions")
    def completions(req: CompletionRequest):
        t0 = time.time()
        app.state.requests += 1
        prompts = [req.prompt] if isinstance(req.prompt, str) else req.prompt
        if not prompts:
            raise HTTPException(400, "empty prompt list")
        texts = llm(prompts, [req.max_tokens] * len(prompts))
        _observe("completions", t0)
        return {"object": "text_completion",
                "choices": [{"index": i, "text": t}
                            for i, t in enumerate(texts)]}

    @app.post("/v1/embeddings")
    def embeddings(req: EmbeddingRequest):
        app.state.requests += 1
        if embedder is None:
            raise HTTPException(503, "no embedder configured")
        t0 = time.time()
        texts = [req.input] if isinstance(req.input, str) else req.input
        vecs = [embedder.embed(t) for t in texts]
        _observe("embeddings", t0)
        return {"object": "list",
                "data": [{"index": i, "embedding": v.tolist()}
                         for i, v in enumerate(vecs)],
                "dims": len(vecs[0]) if vecs else 0}

    @app.post("/v1/search")
    def search(req: SearchRequest):
        app.state.requests += 1
        if index is None:
            raise HTTPException(503, "no vector index configured")
        if isinstance(req.query, str):
            if embedder is None:
                raise HTTPException(503, "no embedder for text queries")
            q = embedder.embed(req.query)
        else:
            import numpy as np
            q = np.asarray(req.query, dtype="float32")
            if q.shape != (index.dim,):
                raise HTTPException(
                    400, f"query dims {q.shape} != index dim {index.dim}")
        t0 = time.time()
        hits = index.search(q, req.k)
        _observe("search", t0)
        return {"hits": [{"document_id": h.document_id, "chunk": h.chunk,
                          "score": h.score, "metadata": h.metadata}
                         for h in hits]}

    @app.post("/v1/agents/{name}")
    def run_agent(name: str, req: AgentRequest):
        app.state.requests += 1
        fn = agents.get(name)
        if fn is None:
            raise HTTPException(404, f"unknown agent {name!r}")
        t0 = time.time()
        res = fn(req.prompt, req.record_key)
        _observe("agents", t0)
        return {"agent": name,
                "status": res.status,
                "response": res.response,
                "iterations": getattr(res, "iterations", None),
                "tool_calls": getattr(res, "tool_calls", None)}

    return app


def build_lab_app(device: str = "cpu", model: str | None = None,
                  labs: tuple[int, ...] = (1, 2)):
    """Wire the app from a lab Deployment: the deployment's LLM (stub on
    CPU, hipGraph engine on 'cuda'), the lab2 document index, and every
    CREATE AGENT in the lab catalogs exposed at /v1/agents/{name} with
    AI_RUN_AGENT semantics (model-driven TOOL_CALL loop against the stub
    MCP server's tool set)."""
    from .agents.mcp import McpClient, StubMcpServer
    from .agents.runner import drive_episode, episode
    from .labs.deploy import Deployment
    from .labs.pipelines import mcp_tool_fn

    dep = Deployment(labs=labs, device=device, model=model)
    if 2 in labs:
        dep.datagen(2)
    server = StubMcpServer().start()
    client = McpClient(server.mcp_endpoint)
    tool_fn = mcp_tool_fn(client)
    # tool schemas -> per-turn grammars: served agents are model-driven
    # (grammar-constrained tool-call decisions, models/grammar.py)
    tool_schemas = {t["name"]: t.get("inputSchema", {})
                    for t in client.tools_list()}
    llm_batch = dep.llm()

    def make_agent_fn(spec):
        def call_llm(p, t, g=None):
            if g is not None and hasattr(llm_batch, "submit_turn"):
                return llm_batch([p], [t], None, [g])[0]   # EngineLLM
            return llm_batch([p], [t])[0]

        def run(prompt: str, record_key=None):
            return drive_episode(
                episode(spec, prompt, tool_schemas=tool_schemas),
                call_llm, tool_fn)
        return run

    agents = {name: make_agent_fn(dep.catalog.agent_spec(name))
              for name in dep.catalog.agents}
    app = create_app(llm_batch, embedder=dep.embedder,
                     index=dep._index(2) if 2 in labs else None,
                     agents=agents,
                     sql_executor=dep.sql_executor(labs[0],
                                                   mcp_server=server))
    app.state.mcp_server = server          # kept alive with the app
    return app


def main(argv=None) -> int:
    import argparse

    import uvicorn
    p = argparse.ArgumentParser(prog="qsa-serve")
    p.add_argument("--host", default="127.0.0.1")
    p.add_argument("--port", type=int, default=8080)
    p.add_argument("--device", default="cpu")
    p.add_argument("--model", default=None)
    args = p.parse_args(argv)
    app = build_lab_app(device=args.device, model=args.model)
    uvicorn.run(app, host=args.host, port=args.port)
    return 0


if __name__ == "__main__":
    raise SystemExit(main())
