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
