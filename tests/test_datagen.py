"""Lab datagen determinism contracts."""

from quickstart_streaming_agents_amd.labs import datagen, schemas
from quickstart_streaming_agents_amd.runtime.anomaly import AnomalyDetector
from quickstart_streaming_agents_amd.runtime.windows import TumblingWindows, aggregate
from quickstart_streaming_agents_amd.wire import Broker

MIN5 = 5 * 60 * 1000
H6 = 6 * 3600 * 1000


def test_lab1_shapes_and_determinism():
    c1, c2 = datagen.lab1_customers(42), datagen.lab1_customers(42)
    assert c1 == c2 and len(c1) == 50
    assert len({c["customer_id"] for c in c1}) == 50
    products = datagen.lab1_products()
    assert len(products) == 17
    orders = datagen.lab1_orders(42, 10, end_ms=10_000_000)
    assert len(orders) == 10
    assert orders[-1]["order_ts"] == 10_000_000
    assert orders[1]["order_ts"] - orders[0]["order_ts"] == 30_000
    pids = {p["product_id"] for p in products}
    assert all(o["product_id"] in pids for o in orders)


def test_lab1_publish_order_and_purge():
    b = Broker()
    datagen.publish_lab1(b, n_orders=10)
    assert b.topic("customers").message_count() == 50
    assert b.topic("products").message_count() == 17
    assert b.topic("orders").message_count() == 10
    datagen.publish_lab1(b, n_orders=5)  # purge-then-publish is idempotent
    assert b.topic("orders").message_count() == 5


def test_lab3_volume_span_and_sort():
    rides = datagen.lab3_ride_requests(42, 288)
    assert len(rides) >= 28_000  # test_lab3.py:188-221 contract
    ts = [r["request_ts"] for r in rides]
    assert ts == sorted(ts)
    assert max(ts) < 288 * MIN5
    assert min(ts) >= 0
    zones = {r["pickup_zone"] for r in rides}
    assert zones == set(schemas.LAB3_ZONES)
    assert all(1 <= r["number_of_passengers"] <= 5 for r in rides)
    assert all(50.0 <= r["price"] <= 150.0 for r in rides)


def test_lab3_surge_contract_end_to_end_anomaly():
    """The full determinism contract: windows + detector flag ONLY the
    French Quarter final-window surge (test_lab3.py:248-257)."""
    rides = datagen.lab3_ride_requests(42, 288)
    tw = TumblingWindows(MIN5, lambda r: r["pickup_zone"],
                         lambda r: r["request_ts"], watermark_delay_ms=5000)
    panes = tw.feed(rides) + tw.flush()
    rows = aggregate(panes, {"request_count": len})
    rows.sort(key=lambda r: (r["window_start"], r["key"]))
    det = AnomalyDetector(min_training_size=286, max_training_size=7000,
                          confidence_percentage=99.9, enable_stl=False)
    anomalies = []
    for r in rows:
        res = det.update(r["key"], float(r["request_count"]))
        if res.is_anomaly and r["request_count"] > res.upper_bound:
            anomalies.append(r)
    assert 1 <= len(anomalies) <= 2
    assert {a["key"] for a in anomalies} == {"French Quarter"}


def test_lab4_volume_and_naples_contract():
    claims = datagen.lab4_claims(42, 14, per_window=88)
    assert len(claims) >= 33_000  # test_lab4.py:215-237 contract
    ts = [c["claim_timestamp"] for c in claims]
    assert ts == sorted(ts)
    assert {c["city"] for c in claims} == set(schemas.LAB4_CITIES)

    tw = TumblingWindows(H6, lambda r: r["city"],
                         lambda r: r["claim_timestamp"], watermark_delay_ms=5000)
    panes = tw.feed(claims) + tw.flush()
    rows = aggregate(panes, {
        "total_claim_amount": lambda rs: sum(float(r["claim_amount"]) for r in rs),
        "claim_count": len,
    })
    rows.sort(key=lambda r: (r["window_start"], r["key"]))
    det = AnomalyDetector(min_training_size=8, max_training_size=50,
                          confidence_percentage=95.0, enable_stl=False)
    anomalies = []
    for r in rows:
        res = det.update(r["key"], r["total_claim_amount"])
        if res.is_anomaly and r["total_claim_amount"] > res.upper_bound:
            anomalies.append(r)
    # Naples spikes in the final 2 days; only Naples is flagged high.
    assert len(anomalies) >= 1
    assert {a["key"] for a in anomalies} == {"Naples"}
    # first flagged window is in the last 2 days (8 windows) of the stream
    n_windows = 14 * 4
    assert anomalies[0]["window_start"] >= (n_windows - 8) * H6


def test_lab2_documents_schema():
    docs = datagen.lab2_documents(42, 64)
    assert len(docs) == 64
    assert len({d["document_id"] for d in docs}) == 64
    assert all(d["char_count"] == len(d["chunk"]) for d in docs)
    b = Broker()
    datagen.publish_lab2(b)
    assert b.topic("documents").message_count() == 64
    assert b.topic("queries").message_count() == 1
