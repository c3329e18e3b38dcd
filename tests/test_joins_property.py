"""Property-based join tests: TTLTable eviction vs a brute-force rule,
interval_join vs an O(n*m) reference (LAB4-Walkthrough.md:207-238 claim
BETWEEN window_time - 6h AND window_time semantics)."""

from hypothesis import given, settings
from hypothesis import strategies as st

from quickstart_streaming_agents_amd.runtime.joins import (TTLTable,
                                                           interval_join)

TTL = 1000


@settings(max_examples=200, deadline=None)
@given(st.lists(st.tuples(
    st.sampled_from(["put", "get"]),
    st.sampled_from(["k1", "k2", "k3"]),
    st.integers(min_value=0, max_value=10_000)), max_size=80))
def test_ttl_table_matches_brute_force(ops):
    """Monotone-or-not timestamps: get(k, now) returns the latest upsert
    of k iff now - upsert_ts <= TTL."""
    table = TTLTable(key_fn=lambda r: r["k"], ttl_ms=TTL)
    shadow = {}
    for op, key, ts in ops:
        if op == "put":
            table.upsert({"k": key, "ts": ts}, ts)
            shadow[key] = ts
        else:
            got = table.get(key, ts)
            want_ts = shadow.get(key)
            if want_ts is not None and ts - want_ts <= TTL:
                assert got is not None and got["ts"] == want_ts
            elif want_ts is None or ts - want_ts > TTL:
                # expired or never inserted -> None (expired also deletes)
                assert got is None
                if want_ts is not None and ts - want_ts > TTL:
                    del shadow[key]


@settings(max_examples=200, deadline=None)
@given(
    st.lists(st.tuples(st.integers(0, 5000), st.sampled_from("xy")),
             max_size=40),
    st.lists(st.tuples(st.integers(0, 5000), st.sampled_from("xy")),
             max_size=40),
    st.integers(-2000, 0), st.integers(0, 2000))
def test_interval_join_matches_nested_loop(lefts, rights, lo, hi):
    L = [{"lts": t, "k": k, "li": i} for i, (t, k) in enumerate(lefts)]
    R = [{"rts": t, "k": k, "ri": i} for i, (t, k) in enumerate(rights)]
    got = interval_join(L, R,
                        left_ts=lambda r: r["lts"],
                        right_ts=lambda r: r["rts"],
                        key_left=lambda r: r["k"],
                        key_right=lambda r: r["k"],
                        lower_ms=lo, upper_ms=hi)
    want = sorted((l["li"], r["ri"]) for l in L for r in R
                  if l["k"] == r["k"]
                  and r["rts"] + lo <= l["lts"] <= r["rts"] + hi)
    assert sorted((m["li"], m["ri"]) for m in got) == want
    # left columns win on collision (claim row overrides anomaly row)
    for m in got:
        assert m["lts"] == L[m["li"]]["lts"]


def test_enrich_join_columnar_matches_dict_reference():
    """The K8 columnar join path (CPU fallback of the GPU hash table)
    must match enrich_join/TTLTable on lab1-shaped data, including
    duplicate keys (latest event time wins) and TTL expiry."""
    from quickstart_streaming_agents_amd.runtime.joins import (
        TTLTable, enrich_join, enrich_join_columnar)
    H = 3_600_000
    custs = [
        {"customer_id": "c1", "customer_email": "old@x.com", "updated_at": 10},
        {"customer_id": "c1", "customer_email": "new@x.com", "updated_at": 20},
        {"customer_id": "c2", "customer_email": "b@x.com", "updated_at": 15},
        {"customer_id": "c3", "customer_email": "stale@x.com", "updated_at": 0},
    ]
    prods = [{"product_id": "p1", "product_name": "widget", "updated_at": 5}]
    orders = [
        {"order_id": "o1", "customer_id": "c1", "product_id": "p1",
         "order_ts": 100},
        {"order_id": "o2", "customer_id": "c2", "product_id": "p1",
         "order_ts": 200},
        {"order_id": "o3", "customer_id": "cX", "product_id": "p1",
         "order_ts": 300},                      # missing dim -> held back
        {"order_id": "o4", "customer_id": "c3", "product_id": "p1",
         "order_ts": H + 1},                    # c3 expired at this ts
    ]
    ct = TTLTable(lambda r: r["customer_id"], ttl_ms=H)
    pt = TTLTable(lambda r: r["product_id"], ttl_ms=H)
    for c in custs:
        ct.upsert(c, c["updated_at"])
    for p in prods:
        pt.upsert(p, p["updated_at"])
    ref = enrich_join(orders, lambda r: r["order_ts"],
                      [(ct, lambda r: r["customer_id"], None),
                       (pt, lambda r: r["product_id"], None)])
    got = enrich_join_columnar(
        orders, lambda r: r["order_ts"],
        [(custs, "customer_id", "updated_at", "customer_id", H),
         (prods, "product_id", "updated_at", "product_id", H)])
    assert got == ref
    assert [r["order_id"] for r in got] == ["o1", "o2"]
    assert got[0]["customer_email"] == "new@x.com"  # latest ts wins


def test_lab1_enriched_orders_gpu_flag_matches_cpu():
    """lab1_enriched_orders(use_gpu=True) on CPU fallback == dict path."""
    from quickstart_streaming_agents_amd.labs import datagen, pipelines
    from quickstart_streaming_agents_amd.wire import Broker
    b1, b2 = Broker(), Broker()
    datagen.publish_lab1(b1)
    datagen.publish_lab1(b2)
    a = pipelines.lab1_enriched_orders(b1, use_gpu=False)
    b = pipelines.lab1_enriched_orders(b2, use_gpu=True)  # CPU fallback
    assert a == b and len(a) >= 1


def test_streaming_join_random_interleavings_match_batch():
    """Property: for ANY arrival interleaving of the three lab1-shaped
    topics (including duplicate sends), the streaming two-sided join
    cascade emits exactly the matched combinations a batch join of the
    same multiset produces — each pair exactly once (the r1 advisor bug
    class)."""
    import itertools
    import random

    from quickstart_streaming_agents_amd.sql.catalog import Catalog
    from quickstart_streaming_agents_amd.sql.exec import SqlExecutor
    from quickstart_streaming_agents_amd.sql.stream import StreamingPipeline
    from quickstart_streaming_agents_amd.wire import Broker

    rng = random.Random(11)
    for trial in range(8):
        orders = [{"o": f"o{i}", "c": f"c{rng.randrange(3)}",
                   "p": f"p{rng.randrange(3)}", "ts": i}
                  for i in range(rng.randrange(1, 7))]
        custs = [{"c": f"c{i}", "cv": f"C{i}-{trial}"} for i in range(3)]
        prods = [{"p": f"p{i}", "pv": f"P{i}-{trial}"} for i in range(3)]
        events = [("orders", o) for o in orders] + \
                 [("custs", c) for c in custs] + \
                 [("prods", p) for p in prods]
        # duplicate one dimension record (a second copy must re-match)
        dup = rng.choice(custs)
        events.append(("custs", dict(dup)))
        rng.shuffle(events)

        cat = Catalog()
        cat.execute("""
        CREATE TABLE joined AS
          SELECT o.o AS o, c.cv AS cv, p.pv AS pv
          FROM orders o JOIN custs c ON o.c = c.c
                        JOIN prods p ON o.p = p.p;
        """)
        broker = Broker()
        for t in ("orders", "custs", "prods"):
            broker.create_topic(t)
        pipe = StreamingPipeline(SqlExecutor(cat, broker))
        got = []
        for topic, rec in events:
            broker.topics[topic].append(dict(rec), partition=0)
            if rng.random() < 0.6:           # advance at random points
                got += pipe.advance()["joined"]
        got += pipe.advance()["joined"]

        # batch reference over the multiset: every (order, cust-copy,
        # prod) key-match combination appears exactly once
        n_copies = {}
        for t, r in events:
            if t == "custs":
                n_copies[r["c"]] = n_copies.get(r["c"], 0) + 1
        want = []
        for o in orders:
            for c in custs:
                if c["c"] != o["c"]:
                    continue
                for _ in range(n_copies.get(c["c"], 1) if c["c"] == dup["c"]
                               else 1):
                    for p in prods:
                        if p["p"] == o["p"]:
                            want.append((o["o"], c["cv"], p["pv"]))
        assert sorted((r["o"], r["cv"], r["pv"]) for r in got) == \
            sorted(want), f"trial {trial}"
