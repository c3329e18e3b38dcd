#!/usr/bin/env python3
"""Config lab4 single-GPU E2E: 6-h windows over ~30k claims -> Naples
anomaly -> interval join -> policy RAG -> fraud verdict agent (no tools),
with the real Llama-3-8B engine."""
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

from quickstart_streaming_agents_amd.labs.deploy import Deployment


def main():
    model = sys.argv[1] if len(sys.argv) > 1 else "llama3-8b"
    dep = Deployment(labs=(4,), device="cuda:0", model=model)
    t0 = time.perf_counter()
    dep.datagen(4)
    t_ingest = time.perf_counter() - t0
    dep.llm()
    t0 = time.perf_counter()
    rows = dep.run(4)
    dt = time.perf_counter() - t0
    n = dep.broker.topic("claims").message_count()
    print(f"ingest: {n} claims in {t_ingest:.2f}s ({n / t_ingest:,.0f} rec/s)")
    print(f"pipeline (window+anomaly+join+RAG+verdict agent): {dt:.2f}s "
          f"for {len(rows)} claims reviewed")
    allowed = {"APPROVE", "APPROVE_PARTIAL", "REQUEST_DOCS",
               "DENY_INELIGIBLE", "DENY_FRAUD"}
    for r in rows[:3]:
        print(f"  {r['claim_id']} city={r['city']} verdict={r['verdict']}")
    assert all(r["city"] == "Naples" for r in rows)
    assert all(r["verdict"] in allowed for r in rows)
    print("lab4 contracts ok")


if __name__ == "__main__":
    main()
