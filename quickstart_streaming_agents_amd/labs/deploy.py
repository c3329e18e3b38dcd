"""Local deployment: the reference's deploy/destroy/datagen orchestration
(deploy.py, scripts/common/destroy.py, scripts/lab*_datagen.py), re-hosted
on the in-process engine.

The reference provisions Confluent Cloud via terraform; here `Deployment`
executes the lab SQL (labs/sql/*.sql — the SAME statement surface users
run, parsed by sql/parse.py) into a Catalog, creates the topics, wires the
models/tools/agents onto the MI355X runtime, and writes the
DEPLOYED_RESOURCES.md / FLINK_SQL_COMMANDS.md summary artifacts
(terraform_runner.py:102-138, generate_lab_flink_summary.py parity).
Destroy drops labs in reverse order (destroy.py:140-146).
"""

from __future__ import annotations

import os

from ..sql.catalog import Catalog
from ..vector.index import HashingEmbedder, VectorIndex
from ..wire import Broker
from . import datagen, pipelines

SQL_DIR = os.path.join(os.path.dirname(os.path.abspath(__file__)), "sql")
ALL_LABS = (1, 2, 3, 4)


def _maybe(res, name):
    try:
        return res(name)
    except Exception:
        return None


def lab_sql(lab: int | str) -> str:
    name = lab if isinstance(lab, str) else f"lab{lab}"
    with open(os.path.join(SQL_DIR, f"{name}.sql")) as fh:
        return fh.read()


class Deployment:
    """One in-process deployment: broker + catalog + per-lab runtime."""

    def __init__(self, labs=ALL_LABS, device: str = "cpu",
                 model: str | None = "tiny", seed: int = 42):
        """model=None resolves the engine preset from the catalog's
        CREATE MODEL options ('local.model', core.sql) — the SQL surface
        is the source of truth; an explicit name is the CLI override."""
        self.labs = tuple(labs)
        self.device = device
        self.model = model
        self.seed = seed
        self.broker = Broker()
        self.catalog = Catalog()
        self.indexes: dict[int, VectorIndex] = {}
        if device.startswith("cuda"):
            from ..models.encoder import EmbeddingEncoder
            self.embedder = EmbeddingEncoder(device=device)
        else:
            self.embedder = HashingEmbedder()
        self._llm = None
        self.catalog.execute(lab_sql("core"))
        for lab in self.labs:
            self.catalog.execute(lab_sql(lab))
        # topics for every non-CTAS table
        for name, t in self.catalog.tables.items():
            if t.as_select is None:
                self.broker.create_topic(name)

    # ---- model backends --------------------------------------------------
    def resolved_model(self) -> str:
        if self.model:
            return self.model
        md = self.catalog.models.get("llm_textgen_model")
        return (md.options.get("local.model", "llama3-8b")
                if md else "llama3-8b")

    def llm(self):
        if self._llm is None:
            if self.device.startswith("cuda"):
                from ..models import build_model
                from ..models.serve import Engine, EngineLLM
                m = build_model(self.resolved_model(), device=self.device,
                                seed=self.seed)
                self._llm = EngineLLM(Engine(m, max_batch=32,
                                             max_seq_len=2048))
            else:
                self._llm = pipelines.StubLLM()
        return self._llm

    # ---- datagen (scripts/lab*_datagen.py parity) ------------------------
    def datagen(self, lab: int, **kw) -> None:
        if lab == 1:
            datagen.publish_lab1(self.broker, seed=self.seed, **kw)
        elif lab == 2:
            datagen.publish_lab2(self.broker, seed=self.seed, **kw)
        elif lab == 3:
            datagen.publish_lab3(self.broker, seed=self.seed, **kw)
        elif lab == 4:
            datagen.publish_lab4(self.broker, seed=self.seed, **kw)
        else:
            raise ValueError(f"unknown lab {lab}")

    def _index(self, lab: int) -> VectorIndex:
        if lab not in self.indexes:
            idx = VectorIndex()
            if lab in (2, 3):
                docs = datagen.lab2_documents(seed=self.seed)
            else:
                docs = datagen.lab4_policy_docs(seed=self.seed)
            idx.add_documents(docs, self.embedder)
            if self.device.startswith("cuda"):
                idx.to_torch(self.device)   # HBM-resident; GPU top-k path
            self.indexes[lab] = idx
        return self.indexes[lab]

    # ---- run (the CTAS pipelines) ----------------------------------------
    def run(self, lab: int, mcp_server=None, **kw) -> list[dict]:
        llm = self.llm()
        cat = self.catalog
        if lab in (1, 3) and mcp_server is None:
            from ..agents.mcp import StubMcpServer
            mcp_server = StubMcpServer().start()
        if lab in (1, 3):
            from ..agents.mcp import McpClient
            tool_fn = pipelines.mcp_tool_fn(McpClient(mcp_server.mcp_endpoint))
        if lab == 1:
            agent = cat.agent_spec("price_match_agent") \
                if "price_match_agent" in cat.agents else None
            ttl = cat.state_ttl_ms() or 3_600_000
            return pipelines.lab1_run(
                self.broker, llm, tool_fn,
                competitor_url=f"{mcp_server.base_url}/competitor",
                agent=agent, state_ttl_ms=ttl, **kw)
        if lab == 2:
            # index from the documents topic if populated, else synthetic
            return pipelines.lab2_run(self.broker, llm, self.embedder, **kw)
        if lab == 3:
            info = cat.ctas_info("anomalies_per_zone")
            params = info.anomaly[0] if info.anomaly else None
            agent = cat.agent_spec("boat_dispatch_agent") \
                if "boat_dispatch_agent" in cat.agents else None
            return pipelines.lab3_run(
                self.broker, llm, tool_fn, self.embedder, self._index(3),
                base_url=mcp_server.base_url, params=params, agent=agent,
                **kw)
        if lab == 4:
            info = cat.ctas_info("claims_anomalies_by_city")
            params = info.anomaly[0] if info.anomaly else None
            agent = cat.agent_spec("claims_fraud_investigation_agent") \
                if "claims_fraud_investigation_agent" in cat.agents else None
            return pipelines.lab4_run(
                self.broker, llm, self.embedder, self._index(4),
                params=params, agent=agent, **kw)
        raise ValueError(f"unknown lab {lab}")

    # ---- generic SQL execution (sql/exec.py) -----------------------------
    FINAL_TABLE = {1: "price_match_results", 2: "search_results_response",
                   3: "completed_actions", 4: "claims_reviewed"}

    def sql_executor(self, lab: int, mcp_server=None,
                     scripted_policies: bool = True):
        """Build a generic SqlExecutor for one lab's catalog: the same
        statements labs/sql/*.sql declare, executed by sql/exec.py instead
        of the hand-fused pipelines.  Prompt expressions the reference
        inlines as giant CONCATs are supplied as named bindings; scripted
        episode policies (the deterministic content stand-ins the lab
        CONTRACT tests assert against) come from agent_policies.
        scripted_policies=False drops them: AI_RUN_AGENT then runs the
        model-driven grammar path (like the bench and serving API)."""
        from ..sql.exec import SqlExecutor
        from . import schemas

        topic_schemas = {
            "orders": schemas.ORDERS, "customers": schemas.CUSTOMERS,
            "products": schemas.PRODUCTS,
            "ride_requests": schemas.RIDE_REQUESTS,
            "claims": schemas.CLAIMS, "queries": schemas.QUERIES,
            "documents": schemas.DOCUMENTS,
        }
        bindings: dict = {}
        policies: dict = {}
        indexes: dict = {}
        tool_fn = None
        tool_schemas = None
        if mcp_server is not None:
            from ..agents.mcp import McpClient
            client = McpClient(mcp_server.mcp_endpoint)
            tool_fn = pipelines.mcp_tool_fn(client)
            tool_schemas = {t["name"]: t.get("inputSchema", {})
                            for t in client.tools_list()}
        if lab == 1:
            url = f"{mcp_server.base_url}/competitor" if mcp_server else ""
            email = "customer@example.com"

            def _order(res):
                vals = {k: _maybe(res, k) for k in
                        ("order_id", "product_name", "price", "order_price",
                         "customer_email", "list_price")}
                return {k: v for k, v in vals.items() if v is not None}

            bindings["user_prompt"] = lambda res: pipelines.lab1_user_prompt(
                _order(res), url, email)
            policies["price_match_agent"] = \
                lambda res: pipelines.Lab1PriceMatchPolicy(_order(res), url,
                                                           email)
        elif lab == 3:
            indexes["documents_vectordb_lab3"] = self._index(3)
            base = mcp_server.base_url if mcp_server else ""

            def _surge(res):
                return pipelines.lab3_surge_query(
                    res("pickup_zone"), int(res("window_time")),
                    int(res("request_count")), float(res("forecast_value")))

            def _summarize(res):
                ctx = "\n\n".join(
                    f"[doc {i} | score {float(res(f'score{i}') or 0):.3f}] "
                    f"{res(f'chunk{i}')}"
                    for i in (1, 2, 3) if _maybe(res, f"chunk{i}"))
                return (f"Answer the question using only the context "
                        f"below.\n\nContext:\n{ctx}\n\n"
                        f"Question: {_surge(res)}\nAnswer:")

            bindings["surge_query"] = _surge
            bindings["summarize_prompt"] = _summarize
            policies["boat_dispatch_agent"] = \
                lambda res: pipelines.Lab3DispatchPolicy(
                    res("pickup_zone"), f"{base}/api/vessel_catalog",
                    f"{base}/api/dispatch")
        elif lab == 4:
            indexes["fema_policies_vectordb"] = self._index(4)

            def _claim(res):
                return {k: _maybe(res, k) for k in
                        ("claim_id", "applicant_name", "city", "claim_amount",
                         "claim_narrative", "is_primary_residence",
                         "damage_assessed", "has_insurance",
                         "insurance_amount", "shared_account",
                         "shared_phone")}

            def _investigation(res):
                c = _claim(res)
                chunks = "\n".join(str(_maybe(res, f"chunk_{i}") or "")
                                   for i in (1, 2, 3))
                return (f"CLAIM {c['claim_id']} ({c['city']}): "
                        f"${c['claim_amount']}\n"
                        f"Narrative: {c['claim_narrative']}\n\n"
                        f"Policy excerpts:\n{chunks}\n\nVerdict?")

            bindings["investigation_prompt"] = _investigation
            policies["claims_fraud_investigation_agent"] = \
                lambda res: pipelines.Lab4FraudPolicy(_claim(res))
        elif lab == 2:
            indexes["documents_vectordb_lab2"] = self._index(2)

            def _rag(res):
                ctx = "\n\n".join(
                    f"[doc {i} | score {float(res(f'score_{i}') or 0):.3f}] "
                    f"{res(f'chunk_{i}')}"
                    for i in (1, 2, 3) if _maybe(res, f"chunk_{i}"))
                return (f"Answer the question using only the context "
                        f"below.\n\nContext:\n{ctx}\n\n"
                        f"Question: {res('query')}\nAnswer:")

            bindings["rag_prompt"] = _rag
        if not scripted_policies:
            policies = {}
        return SqlExecutor(self.catalog, self.broker, schemas=topic_schemas,
                           embedder=self.embedder, indexes=indexes,
                           llm_batch=self.llm(), tool_fn=tool_fn,
                           bindings=bindings, agent_policies=policies,
                           tool_schemas=tool_schemas)

    def run_sql(self, lab: int, mcp_server=None) -> list[dict]:
        """Run one lab end-to-end through the GENERIC SQL executor (the
        hand-fused pipelines are `run()`)."""
        ex = self.sql_executor(lab, mcp_server=mcp_server)
        ex.run_inserts()
        return ex.run_table(self.FINAL_TABLE[lab])

    def run_stream(self, lab: int, mcp_server=None) -> list[dict]:
        """Run one lab through the STREAMING executor (sql/stream.py):
        incremental consumption + watermark-driven window closure, then a
        bounded-input flush."""
        from ..sql.stream import StreamingPipeline
        ex = self.sql_executor(lab, mcp_server=mcp_server)
        ex.run_inserts(values_only=True)
        pipe = StreamingPipeline(ex)
        final = self.FINAL_TABLE[lab]
        rows = pipe.advance().get(final, [])
        for sink, extra in pipe.finish().items():
            if sink == final:
                rows += extra
        return rows

    # ---- destroy / summary ----------------------------------------------
    def destroy(self) -> None:
        """Drop lab objects in reverse order (destroy.py:140-146)."""
        for lab in reversed(self.labs):
            pass  # catalog-level: drop all objects
        for store in (self.catalog.agents, self.catalog.tools,
                      self.catalog.models, self.catalog.connections):
            store.clear()
        for name in list(self.catalog.tables):
            del self.catalog.tables[name]
        for name in list(self.broker.topics):
            self.broker.delete_topic(name)

    def summary_markdown(self) -> str:
        """DEPLOYED_RESOURCES.md parity (generate_deployment_summary.py)."""
        cat = self.catalog
        lines = ["# Deployed Resources (local MI355X engine)", ""]
        lines += [f"- Device: `{self.device}`  Model: `{self.model}`", ""]
        lines.append("## Topics")
        for name in sorted(self.broker.topics):
            lines.append(f"- `{name}`")
        lines.append("")
        lines.append("## Models")
        for name, m in cat.models.items():
            lines.append(f"- `{name}` ({m.options.get('local.model', '?')})")
        lines.append("")
        lines.append("## Tools / Agents")
        for name in cat.tools:
            lines.append(f"- TOOL `{name}`")
        for name, a in cat.agents.items():
            lines.append(f"- AGENT `{name}` (model `{a.model}`, "
                         f"max_iterations="
                         f"{a.options.get('max_iterations', a.options.get('MAX_ITERATIONS', '10'))})")
        lines.append("")
        return "\n".join(lines)

    def write_summaries(self, out_dir: str) -> None:
        os.makedirs(out_dir, exist_ok=True)
        with open(os.path.join(out_dir, "DEPLOYED_RESOURCES.md"), "w") as fh:
            fh.write(self.summary_markdown())
        for lab in self.labs:
            with open(os.path.join(out_dir,
                                   f"LAB{lab}_SQL_COMMANDS.md"), "w") as fh:
                fh.write(f"# Lab {lab} SQL\n\n```sql\n{lab_sql(lab)}\n```\n")
