"""CLI entry points — the reference's `uv run deploy|destroy|*_datagen|tests`
surface (pyproject.toml:90-165) on the local MI355X engine.

    python -m quickstart_streaming_agents_amd deploy  [--labs 1,3] [--dir out]
    python -m quickstart_streaming_agents_amd run     --lab 1 [--device cuda]
    python -m quickstart_streaming_agents_amd datagen --lab 3
    python -m quickstart_streaming_agents_amd validate
    python -m quickstart_streaming_agents_amd destroy --dir out
"""

from __future__ import annotations

import argparse
import json
import sys


def _labs(arg: str):
    return tuple(int(x) for x in arg.split(",") if x.strip())


def cmd_deploy(args) -> int:
    from .labs.deploy import Deployment
    dep = Deployment(labs=_labs(args.labs), device=args.device,
                     model=args.model)
    dep.write_summaries(args.dir)
    print(dep.summary_markdown())
    print(f"[deploy] summaries written to {args.dir}/")
    return 0


def cmd_run(args) -> int:
    from .agents.mcp import StubMcpServer
    from .labs.deploy import Deployment
    dep = Deployment(labs=(args.lab,), device=args.device, model=args.model)
    dep.datagen(args.lab)
    server = StubMcpServer().start()
    try:
        if args.stream:
            rows = dep.run_stream(args.lab, mcp_server=server)
        elif args.sql:
            rows = dep.run_sql(args.lab, mcp_server=server)
        else:
            rows = dep.run(args.lab, mcp_server=server)
    finally:
        server.stop()
    for r in rows[: args.max_print]:
        print(json.dumps(r, default=str)[:400])
    print(f"[run] lab{args.lab}: {len(rows)} output records")
    return 0 if rows else 1


def cmd_datagen(args) -> int:
    from .labs.deploy import Deployment
    dep = Deployment(labs=(args.lab,), device="cpu")
    dep.datagen(args.lab)
    for name, t in sorted(dep.broker.topics.items()):
        n = t.message_count()
        if n:
            print(f"[datagen] {name}: {n} records")
    return 0


def cmd_validate(args) -> int:
    """Advisory checks (scripts/common/validate.py parity): embedding dims
    contract, vector index health, MCP stub round trip, HIP extension."""
    ok = True
    from .vector.index import EMBED_DIM, HashingEmbedder, VectorIndex
    emb = HashingEmbedder()
    v = emb.embed("validate")
    print(f"[validate] embedding dims = {len(v)} "
          f"({'OK' if len(v) == EMBED_DIM else 'FAIL: expected 1536'})")
    ok &= len(v) == EMBED_DIM
    idx = VectorIndex()
    idx.add("d1", "hello world", v)
    hits = idx.search(v, 1)
    print(f"[validate] vector index cosine self-hit = {hits[0].score:.3f} "
          f"({'OK' if hits[0].score > 0.99 else 'FAIL'})")
    ok &= hits[0].score > 0.99
    from .agents.mcp import McpClient, StubMcpServer
    srv = StubMcpServer().start()
    try:
        tools = McpClient(srv.mcp_endpoint).tools_list()
        names = sorted(t["name"] for t in tools)
        print(f"[validate] MCP tools/list -> {names}")
        ok &= "http_get" in names
    finally:
        srv.stop()
    from .sql.catalog import Catalog
    from .sql.exec import SqlExecutor
    from .wire import Broker
    cat = Catalog()
    cat.execute("CREATE TABLE t (a STRING); "
                "CREATE TABLE u AS SELECT s.a FROM t s WHERE s.a <> '';")
    b = Broker()
    b.create_topic("t").append({"a": "x"}, partition=0)
    rows = SqlExecutor(cat, b).run_table("u")
    print(f"[validate] SQL executor round trip: "
          f"{'OK' if rows == [{'a': 'x'}] else 'FAIL'}")
    ok &= rows == [{"a": "x"}]
    from .ops import have_ext
    print(f"[validate] qsa_hip extension importable: {have_ext()}")
    import torch
    print(f"[validate] torch.cuda.is_available(): "
          f"{torch.cuda.is_available()}")
    return 0 if ok else 1


def cmd_destroy(args) -> int:
    import os
    removed = []
    for f in ("DEPLOYED_RESOURCES.md",) + tuple(
            f"LAB{i}_SQL_COMMANDS.md" for i in (1, 2, 3, 4)):
        p = os.path.join(args.dir, f)
        if os.path.exists(p):
            os.unlink(p)
            removed.append(f)
    print(f"[destroy] removed {len(removed)} artifacts from {args.dir}/")
    return 0


def cmd_capture(args) -> int:
    """Capture a lab topic to JSONL (scripts/capture_lab{1,3}_data.py
    parity): datagen then dump base64 wire payloads for replay."""
    from .labs import schemas
    from .labs.capture import capture_topic
    from .labs.deploy import Deployment
    dep = Deployment(labs=(args.lab,), device="cpu")
    dep.datagen(args.lab)
    schema = {1: schemas.ORDERS, 2: schemas.QUERIES,
              3: schemas.RIDE_REQUESTS, 4: schemas.CLAIMS}[args.lab]
    topic = args.topic or {1: "orders", 2: "queries",
                           3: "ride_requests", 4: "claims"}[args.lab]
    n = capture_topic(dep.broker, topic, schema, args.out)
    print(f"[capture] {topic}: {n} records -> {args.out}")
    return 0 if n else 1


def cmd_sql(args) -> int:
    """Execute a .sql file (the lab grammar, docs/SQL.md) against JSONL
    topic data: --data topic=path.jsonl (repeatable), then print the rows
    of --table (or every CTAS).  --interactive reads further statements
    from stdin after the file (if any) is applied."""
    from .sql import parse as P
    from .sql.exec import SqlExecError, SqlExecutor
    server = None
    if args.lab:
        # lab context: catalog + datagen topics + stub-LLM engine wiring,
        # so ML_PREDICT / VECTOR_SEARCH_AGG / AI_RUN_AGENT statements run
        from .agents.mcp import StubMcpServer
        from .labs.deploy import Deployment
        dep = Deployment(labs=(args.lab,), device=args.device)
        dep.datagen(args.lab)
        server = StubMcpServer().start()
        ex = dep.sql_executor(args.lab, mcp_server=server)
        cat, broker = dep.catalog, dep.broker
    else:
        from .sql.catalog import Catalog
        from .wire import Broker
        cat = Catalog()
        broker = Broker()
        ex = SqlExecutor(cat, broker)
    for spec in args.data or []:
        topic_name, _, path = spec.partition("=")
        t = broker.create_topic(topic_name)
        with open(path) as fh:
            for line in fh:
                line = line.strip()
                if line:
                    t.append(json.loads(line), partition=0)

    def run_script(text: str, materialize: bool) -> None:
        for st in P.parse_script(text):
            cat.apply(st)
            if isinstance(st, P.ShowStmt):
                print(f"-- SHOW {st.kind}: {', '.join(cat.show(st.kind))}")
            elif isinstance(st, P.DescribeStmt):
                print(f"-- DESCRIBE {st.name}:")
                for col, ty in cat.describe(st.name):
                    print(f"   {col:32s} {ty}")
            elif isinstance(st, P.ExplainStmt):
                print(f"-- plan for {st.name}:")
                for step in ex.explain(st.name):
                    print(f"   {step}")
            elif isinstance(st, P.InsertInto) and materialize:
                ex.run_inserts()
                cat.inserts.clear()
            elif isinstance(st, P.CreateTable) and st.as_select and \
                    materialize:
                if args.explain:
                    print(f"-- plan for {st.name}:")
                    for step in ex.explain(st.name):
                        print(f"   {step}")
                    continue
                rows = ex.run_table(st.name)
                print(f"-- {st.name}: {len(rows)} rows")
                for r in rows[: args.max_print]:
                    print(json.dumps(r, default=str)[:400])

    if args.file:
        with open(args.file) as fh:
            text = fh.read()
        if args.table:
            cat.execute(text)
            ex.run_inserts()
            rows = ex.run_table(args.table)
            print(f"-- {args.table}: {len(rows)} rows")
            for r in rows[: args.max_print]:
                print(json.dumps(r, default=str)[:400])
        else:
            run_script(text, materialize=True)
    if args.interactive:
        print("-- interactive: end statements with ';' (EOF to quit)")
        buf = []
        for line in sys.stdin:
            buf.append(line)
            if line.rstrip().endswith(";"):
                try:
                    run_script("".join(buf), materialize=True)
                except (ValueError, SqlExecError, KeyError) as e:
                    print(f"-- error: {e}")
                buf = []
    if server is not None:
        server.stop()
    return 0


def cmd_serve(args) -> int:
    from .serve_api import main as serve_main
    return serve_main(["--host", args.host, "--port", str(args.port),
                       "--device", args.device] +
                      (["--model", args.model] if args.model else []))


def cmd_tests(args) -> int:
    import os
    import subprocess
    root = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    cmd = [sys.executable, "-m", "pytest", "-q"]
    if args.quick:
        cmd += ["tests/test_wire.py", "tests/test_sql.py",
                "tests/test_datagen.py"]
    else:
        cmd += ["tests"]
    cmd += ["-m", "gpu" if args.gpu else "not gpu"]
    if args.resume:
        cmd.append("--lf")
    if args.k:
        cmd += ["-k", args.k]
    return subprocess.call(cmd, cwd=root)


def main(argv=None) -> int:
    p = argparse.ArgumentParser(prog="quickstart_streaming_agents_amd")
    sub = p.add_subparsers(dest="cmd", required=True)

    d = sub.add_parser("deploy", help="build catalog + topics, write summaries")
    d.add_argument("--labs", default="1,2,3,4")
    d.add_argument("--device", default="cpu")
    d.add_argument("--model", default=None,
                   help="engine preset; default resolves from the lab "
                        "SQL CREATE MODEL options")
    d.add_argument("--dir", default="deploy_out")
    d.set_defaults(fn=cmd_deploy)

    r = sub.add_parser("run", help="datagen + run one lab end-to-end")
    r.add_argument("--lab", type=int, required=True)
    r.add_argument("--device", default="cpu")
    r.add_argument("--model", default=None,
                   help="engine preset; default resolves from the lab "
                        "SQL CREATE MODEL options (CPU runs use the "
                        "stub LLM regardless)")
    r.add_argument("--max-print", type=int, default=3)
    r.add_argument("--sql", action="store_true",
                   help="run through the generic SQL CTAS executor "
                        "(sql/exec.py) instead of the fused pipelines")
    r.add_argument("--stream", action="store_true",
                   help="run through the incremental streaming executor "
                        "(sql/stream.py)")
    r.set_defaults(fn=cmd_run)

    g = sub.add_parser("datagen", help="publish one lab's synthetic stream")
    g.add_argument("--lab", type=int, required=True)
    g.set_defaults(fn=cmd_datagen)

    v = sub.add_parser("validate", help="advisory environment checks")
    v.set_defaults(fn=cmd_validate)

    x = sub.add_parser("destroy", help="remove deployment artifacts")
    x.add_argument("--dir", default="deploy_out")
    x.set_defaults(fn=cmd_destroy)

    c = sub.add_parser("capture", help="capture a lab topic to JSONL "
                       "(wire payloads, base64) for replay")
    c.add_argument("--lab", type=int, required=True)
    c.add_argument("--topic", default=None)
    c.add_argument("--out", required=True)
    c.set_defaults(fn=cmd_capture)

    q = sub.add_parser("sql", help="execute a .sql file against JSONL "
                       "topic data (docs/SQL.md grammar)")
    q.add_argument("--file", default=None)
    q.add_argument("--lab", type=int, default=None,
                   help="load a lab catalog + datagen topics + stub "
                        "engine so AI statements execute")
    q.add_argument("--device", default="cpu")
    q.add_argument("--interactive", action="store_true",
                   help="read further statements from stdin")
    q.add_argument("--data", action="append",
                   help="topic=path.jsonl (repeatable)")
    q.add_argument("--table", default=None,
                   help="CTAS to materialize (default: all)")
    q.add_argument("--explain", action="store_true",
                   help="print the stage plan instead of executing")
    q.add_argument("--max-print", type=int, default=5)
    q.set_defaults(fn=cmd_sql)

    s = sub.add_parser("serve", help="HTTP serving API "
                       "(completions/embeddings/search/agents)")
    s.add_argument("--host", default="127.0.0.1")
    s.add_argument("--port", type=int, default=8080)
    s.add_argument("--device", default="cpu")
    s.add_argument("--model", default=None)
    s.set_defaults(fn=cmd_serve)

    t = sub.add_parser("tests", help="run the test suite "
                       "(scripts/run_tests.py parity)")
    t.add_argument("--quick", action="store_true",
                   help="fast preflight subset (wire/sql/datagen)")
    t.add_argument("--gpu", action="store_true", help="GPU-marked tests")
    t.add_argument("--resume", action="store_true",
                   help="continue from the last failure (pytest --lf)")
    t.add_argument("-k", default=None, help="pytest -k expression")
    t.set_defaults(fn=cmd_tests)

    args = p.parse_args(argv)
    return args.fn(args)


if __name__ == "__main__":
    sys.exit(main())
