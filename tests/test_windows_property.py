"""Property-based window/watermark tests (hypothesis): random event
streams vs a brute-force reference of Flink TUMBLE semantics
(LAB3-Walkthrough.md:99-133) — every in-watermark row lands in exactly
one pane, panes close exactly once, totals conserve rows."""

from collections import defaultdict

from hypothesis import given, settings
from hypothesis import strategies as st

from quickstart_streaming_agents_amd.runtime.windows import (TumblingWindows,
                                                             aggregate)

SIZE = 1000          # 1 s windows
DELAY = 500


events = st.lists(
    st.tuples(st.integers(min_value=0, max_value=20_000),   # ts
              st.sampled_from(["a", "b", "c"])),            # key
    min_size=0, max_size=200)


def brute_force(rows):
    """Reference: replay the same order with the same late-drop rule,
    then group survivors by (key, window)."""
    panes = defaultdict(list)
    max_ts = -(1 << 62)
    dropped = 0
    for ts, key in rows:
        start = (ts // SIZE) * SIZE
        if start + SIZE <= max_ts - DELAY:
            dropped += 1
            continue
        panes[(key, start)].append(ts)
        max_ts = max(max_ts, ts)
    return panes, dropped


@settings(max_examples=200, deadline=None)
@given(events)
def test_every_surviving_row_in_exactly_one_pane(rows):
    tw = TumblingWindows(SIZE, key_fn=lambda r: r["k"],
                         ts_fn=lambda r: r["ts"],
                         watermark_delay_ms=DELAY)
    closed = []
    for ts, key in rows:                      # one row per feed: worst case
        closed += tw.feed([{"ts": ts, "k": key}])
    closed += tw.flush()

    ref_panes, ref_dropped = brute_force(rows)

    # 1. no pane emitted twice
    seen = set()
    for p in closed:
        assert (p.key, p.window_start) not in seen
        seen.add((p.key, p.window_start))
        # 2. window bounds and membership
        assert p.window_end == p.window_start + SIZE
        for r in p.rows:
            assert p.window_start <= r["ts"] < p.window_end
            assert r["k"] == p.key

    # 3. exact pane-by-pane match with the reference
    got = {(p.key, p.window_start): sorted(r["ts"] for r in p.rows)
           for p in closed}
    want = {(k, s): sorted(v) for (k, s), v in ref_panes.items()}
    assert got == want

    # 4. row conservation: emitted + dropped == fed
    emitted = sum(len(p.rows) for p in closed)
    assert emitted + ref_dropped == len(rows)


@settings(max_examples=100, deadline=None)
@given(events)
def test_batch_vs_single_feed_same_flush_result(rows):
    mk = lambda: TumblingWindows(SIZE, key_fn=lambda r: r["k"],
                                 ts_fn=lambda r: r["ts"],
                                 watermark_delay_ms=DELAY)
    rws = [{"ts": ts, "k": k} for ts, k in rows]
    a = mk(); one = a.feed(rws) + a.flush()
    b = mk()
    two = []
    for r in rws:
        two += b.feed([r])
    two += b.flush()
    key = lambda p: (p.window_start, str(p.key))
    assert [(p.key, p.window_start, sorted(x["ts"] for x in p.rows))
            for p in sorted(one, key=key)] == \
           [(p.key, p.window_start, sorted(x["ts"] for x in p.rows))
            for p in sorted(two, key=key)]


@settings(max_examples=50, deadline=None)
@given(events)
def test_aggregate_count_matches_pane_sizes(rows):
    tw = TumblingWindows(SIZE, key_fn=lambda r: r["k"],
                         ts_fn=lambda r: r["ts"],
                         watermark_delay_ms=DELAY)
    panes = tw.feed([{"ts": ts, "k": k} for ts, k in rows]) + tw.flush()
    out = aggregate(panes, {"request_count": len})
    assert [o["request_count"] for o in out] == [len(p.rows) for p in panes]
    assert all(o["window_time"] == o["window_end"] - 1 for o in out)
