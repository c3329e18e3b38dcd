"""Deterministic synthetic data generators for the four labs.

Reproduces the reference's operating points and determinism contracts
 with seeded generators — no network, no ShadowTraffic:

- lab1: 50 customers / 17 products / orders stream
  (reference scripts/generate_lab1_data.py, seed 42: 50 customers at 216-234,
  17 products at 100-204, orders at 239-256).
- lab3: >=28,000 ride_requests spanning exactly 288 x 5-min windows, 7 NOLA
  zones, surge ONLY in French Quarter in the final window
  (scripts/publish_lab3_data.py:143-170 rebases to 288 windows;
  testing/e2e/test_lab3.py:248-257 asserts <=2 anomalies, French Quarter only).
- lab4: ~36,000 FEMA claims, 14 days, 8 Florida cities, Naples spiking in the
  final 2 days (scripts/lab4_datagen.py; LAB4-Walkthrough.md:66-68;
  test_lab4.py:265-274 asserts the single Naples anomaly).
- lab2: synthetic doc chunks with RAG metadata + sample query
  (scripts/publish_docs.py frontmatter schema; lab2 main.tf:177).
"""

from __future__ import annotations

import random

from . import schemas
from ..wire.topics import AvroProducer, Broker

MIN5_MS = 5 * 60 * 1000
H6_MS = 6 * 3600 * 1000

_FIRST = ["Alex", "Sam", "Jordan", "Taylor", "Casey", "Morgan", "Riley", "Avery",
          "Quinn", "Dana", "Jamie", "Drew", "Cameron", "Skyler", "Reese", "Parker"]
_LAST = ["Lee", "Kim", "Patel", "Garcia", "Nguyen", "Smith", "Johnson", "Brown",
         "Davis", "Martinez", "Lopez", "Wilson", "Clark", "Young", "Hall", "King"]
_STATES = ["CA", "TX", "NY", "FL", "WA", "IL", "MA", "GA", "OH", "CO"]

_PRODUCTS = [
    ("AirPods Pro", 249.00, "Electronics"),
    ("Mechanical Keyboard", 129.99, "Electronics"),
    ("4K Monitor", 399.00, "Electronics"),
    ("Espresso Machine", 549.00, "Kitchen"),
    ("Chef Knife", 89.00, "Kitchen"),
    ("Cast Iron Skillet", 42.50, "Kitchen"),
    ("Running Shoes", 139.95, "Sports"),
    ("Yoga Mat", 34.99, "Sports"),
    ("Carbon Road Bike", 2199.00, "Sports"),
    ("Noise-Cancel Headphones", 329.00, "Electronics"),
    ("Smart Thermostat", 189.00, "Home"),
    ("Robot Vacuum", 479.00, "Home"),
    ("Standing Desk", 599.00, "Office"),
    ("Ergonomic Chair", 749.00, "Office"),
    ("USB-C Dock", 159.00, "Office"),
    ("E-Reader", 139.00, "Electronics"),
    ("Action Camera", 299.00, "Electronics"),
]


# ---------------------------------------------------------------------------
# Lab 1 — orders / customers / products
# ---------------------------------------------------------------------------


def lab1_customers(seed: int = 42, n: int = 50) -> list[dict]:
    rng = random.Random(seed)
    out = []
    for i in range(1, n + 1):
        fn, ln = rng.choice(_FIRST), rng.choice(_LAST)
        out.append({
            "customer_id": f"CUST-{i:03d}",
            "customer_email": f"{fn.lower()}.{ln.lower()}{i}@example.com",
            "customer_name": f"{fn} {ln}",
            "state": rng.choice(_STATES),
            "updated_at": 0,
        })
    return out


def lab1_products(n: int = 17) -> list[dict]:
    out = []
    for i, (name, price, dept) in enumerate(_PRODUCTS[:n], start=1):
        out.append({
            "product_id": f"PROD-{i:03d}",
            "product_name": name,
            "price": price,
            "department": dept,
            "updated_at": 0,
        })
    return out


def lab1_orders(seed: int = 42, n: int = 10, end_ms: int = 0,
                spacing_ms: int = 30_000, n_customers: int = 50,
                n_products: int = 17) -> list[dict]:
    """n orders at fixed spacing ending at end_ms (reference: 30 s spacing,
    ending 5 min before now)."""
    rng = random.Random(seed + 1)
    products = lab1_products(n_products)
    out = []
    start = end_ms - (n - 1) * spacing_ms
    for i in range(n):
        p = products[rng.randrange(n_products)]
        out.append({
            "order_id": f"ORD-{i + 1:05d}",
            "customer_id": f"CUST-{rng.randrange(1, n_customers + 1):03d}",
            "product_id": p["product_id"],
            "price": p["price"],
            "order_ts": start + i * spacing_ms,
        })
    return out


def publish_lab1(broker: Broker, seed: int = 42, n_orders: int = 10,
                 now_ms: int = 10 * 60 * 1000) -> None:
    """Purge-then-publish customers -> products -> orders (reference order:
    publish_lab1_data.py:377-385), orders ending 5 min before now."""
    for topic in ("customers", "products", "orders"):
        broker.create_topic(topic).purge()
    pc = AvroProducer(broker, "customers", schemas.CUSTOMERS)
    for c in lab1_customers(seed):
        c = dict(c, updated_at=now_ms)
        pc.produce(c, key=c["customer_id"], timestamp_ms=now_ms, partition=0)
    pp = AvroProducer(broker, "products", schemas.PRODUCTS)
    for p in lab1_products():
        p = dict(p, updated_at=now_ms)
        pp.produce(p, key=p["product_id"], timestamp_ms=now_ms, partition=0)
    po = AvroProducer(broker, "orders", schemas.ORDERS)
    for o in lab1_orders(seed, n_orders, end_ms=now_ms - 5 * 60 * 1000):
        po.produce(o, key=o["order_id"], timestamp_ms=o["order_ts"], partition=0)


# ---------------------------------------------------------------------------
# Lab 3 — ride_requests (288 x 5-min windows, French Quarter surge)
# ---------------------------------------------------------------------------


def lab3_ride_requests(seed: int = 42, n_windows: int = 288,
                       align_ms: int = 0) -> list[dict]:
    """Deterministic ride stream.

    Steady zones: ~15 requests per 5-min window with small jitter; the surge
    zone (French Quarter) runs steady for windows 0..n-2 then bursts in the
    final window (~6x rate).  Data spans exactly n_windows windows starting at
    align_ms (a 5-min-aligned epoch); sorted chronologically so watermarks
    advance monotonically (publish_lab3_data.py:360-369 sorts by rebased ts).
    """
    rng = random.Random(seed)
    out: list[dict] = []
    rid = 0
    for zone in schemas.LAB3_ZONES:
        t = align_ms + rng.randrange(0, 4000)
        end = align_ms + n_windows * MIN5_MS
        surge_start = align_ms + (n_windows - 1) * MIN5_MS
        while t < end:
            surging = zone == schemas.LAB3_SURGE_ZONE and t >= surge_start
            rid += 1
            out.append({
                "request_id": f"REQ-{rid:06d}",
                "customer_email": f"rider{rng.randrange(1, 2000)}@example.com",
                "pickup_zone": zone,
                "drop_off_zone": rng.choice(
                    [z for z in schemas.LAB3_ZONES if z != zone]),
                "price": round(rng.uniform(50.0, 150.0), 2),
                "number_of_passengers": rng.randint(1, 5),
                "request_ts": t,
            })
            if surging:
                # ~4 s cadence -> ~75 requests in the surge window (~5x steady)
                t += rng.randrange(3600, 4400)
            else:
                # ~20 s cadence, BOUNDED (uniform) jitter: per-window counts
                # stay within the detector's 99.9% band by construction so
                # only the surge is ever flagged (determinism contract).
                t += rng.randrange(17000, 23000)
    out.sort(key=lambda r: r["request_ts"])
    return out


def publish_lab3(broker: Broker, seed: int = 42, n_windows: int = 288,
                 align_ms: int = 0) -> int:
    broker.create_topic("ride_requests").purge()
    prod = AvroProducer(broker, "ride_requests", schemas.RIDE_REQUESTS)
    rides = lab3_ride_requests(seed, n_windows, align_ms)
    for r in rides:
        prod.produce(r, key=r["customer_email"], timestamp_ms=r["request_ts"],
                     partition=0)
    return len(rides)


# ---------------------------------------------------------------------------
# Lab 4 — FEMA claims (14 days, 6-h windows, Naples spike in final 2 days)
# ---------------------------------------------------------------------------

_NARRATIVES = [
    "Flood water entered the ground floor and destroyed flooring, drywall and "
    "two bedrooms of furniture. We evacuated for six days.",
    "Roof partially torn off by hurricane winds; rain damage to kitchen and "
    "living room ceilings. Temporary tarp installed.",
    "Storm surge flooded the garage and utility room, ruining the water heater "
    "and washer/dryer. Mold remediation required.",
    "Fallen tree crushed the carport and damaged the east wall. Debris removal "
    "and structural repair needed.",
    "Wind-driven rain through broken windows saturated carpets and electronics "
    "in the home office.",
    "",
]


def lab4_claims(seed: int = 42, n_days: int = 14, end_ms: int | None = None,
                per_window: int = 88) -> list[dict]:
    """~n_days*4*8*per_window/... claims; ~36k at defaults (8 cities x 56
    6-h windows x ~80).  Non-spike cities decay gently; Naples count and
    amounts spike in the final 2 days."""
    rng = random.Random(seed)
    n_windows = n_days * 4
    if end_ms is None:
        end_ms = n_windows * H6_MS
    start_ms = end_ms - n_windows * H6_MS
    out: list[dict] = []
    cid = 0
    for w in range(n_windows):
        w_start = start_ms + w * H6_MS
        final_2d = w >= n_windows - 8
        for city in schemas.LAB4_CITIES:
            spike = final_2d and city == schemas.LAB4_SPIKE_CITY
            decay = 1.0 - 0.3 * (w / n_windows)
            count = int(per_window * decay * rng.uniform(0.98, 1.02))
            # Per-(city, window) claim totals carry BOUNDED (uniform +-2%)
            # noise so the conf-95 band contains every steady window and the
            # Naples spike (15x) is the single flagged city.
            target_total = per_window * 5000.0 * decay * rng.uniform(0.98, 1.02)
            if spike:
                count = int(per_window * 3.0)
                target_total = per_window * 5000.0 * 15.0 * rng.uniform(0.98, 1.02)
            amounts = [rng.uniform(3000.0, 7000.0) for _ in range(count)]
            scale = target_total / max(sum(amounts), 1e-9)
            for amount in amounts:
                cid += 1
                amount *= scale
                fn, ln = rng.choice(_FIRST), rng.choice(_LAST)
                has_ins = rng.random() < 0.4
                out.append({
                    "claim_id": f"CLM-{cid:06d}",
                    "applicant_name": f"{fn} {ln}",
                    "city": city,
                    "is_primary_residence": "Yes" if rng.random() < 0.8 else "No",
                    "damage_assessed": f"{rng.uniform(1000, 50000):.2f}",
                    "claim_amount": f"{amount:.2f}",
                    "has_insurance": "Yes" if has_ins else "No",
                    "insurance_amount": f"{rng.uniform(0, amount):.2f}" if has_ins else "0",
                    "claim_narrative": rng.choice(_NARRATIVES),
                    "assessment_date": "2024-10-12",
                    "disaster_date": "2024-10-09",
                    "previous_claims_count": str(rng.randrange(0, 4)),
                    "last_claim_date": "2022-08-30" if rng.random() < 0.3 else "",
                    "assessment_source": rng.choice(["inspector", "self-reported"]),
                    "shared_account": "Yes" if rng.random() < 0.05 else "No",
                    "shared_phone": "Yes" if rng.random() < 0.05 else "No",
                    "claim_timestamp": w_start + rng.randrange(0, H6_MS),
                })
    out.sort(key=lambda r: r["claim_timestamp"])
    return out


def publish_lab4(broker: Broker, seed: int = 42, n_days: int = 14,
                 per_window: int = 88) -> int:
    """Purge claims + the 4 downstream topics, then publish chronologically
    (lab4_datagen.py:294-325)."""
    for t in ("claims", "claims_anomalies_by_city", "claims_to_investigate",
              "claims_to_investigate_with_policies", "claims_reviewed"):
        broker.create_topic(t).purge()
    prod = AvroProducer(broker, "claims", schemas.CLAIMS)
    claims = lab4_claims(seed, n_days, per_window=per_window)
    for c in claims:
        prod.produce(c, key=c["claim_id"], timestamp_ms=c["claim_timestamp"],
                     partition=0)
    return len(claims)


# ---------------------------------------------------------------------------
# Lab 2 — documents + queries
# ---------------------------------------------------------------------------

_DOC_TOPICS = [
    ("Creating Tables", "CREATE TABLE defines a dynamic table over a Kafka "
     "topic. Specify columns, a WATERMARK clause for event time, and WITH "
     "options for the connector."),
    ("Tumbling Windows", "TUMBLE assigns each row to a fixed, non-overlapping "
     "window of the given size based on the time attribute DESCRIPTOR."),
    ("ML_PREDICT", "ML_PREDICT invokes a registered remote or local model on "
     "each row, returning the model output columns via LATERAL TABLE."),
    ("Vector Search", "VECTOR_SEARCH_AGG retrieves the top-k nearest document "
     "chunks by cosine similarity between the query embedding and the index."),
    ("Anomaly Detection", "ML_DETECT_ANOMALIES fits a streaming forecaster per "
     "partition and flags points outside the configured confidence band."),
    ("Agents", "CREATE AGENT binds a model, a system prompt and tools; "
     "AI_RUN_AGENT executes the reason-act loop per row with iteration caps."),
    ("Watermarks", "A watermark declares how long to wait for late events; "
     "windows close when the watermark passes their end."),
    ("Joins", "Streaming joins keep keyed state; set sql.state-ttl to bound "
     "state growth for unbounded streams."),
]


def lab2_documents(seed: int = 42, n_chunks: int = 64) -> list[dict]:
    rng = random.Random(seed)
    out = []
    for i in range(n_chunks):
        title, base = _DOC_TOPICS[i % len(_DOC_TOPICS)]
        chunk = f"{base} (section {i // len(_DOC_TOPICS) + 1}; detail code "
        chunk += "".join(rng.choice("abcdefghij") for _ in range(8)) + ")"
        out.append({
            "document_id": f"DOC-{i:04d}",
            "title": title,
            "chunk": chunk,
            "pages": f"{i + 1}-{i + 2}",
            "section_reference": f"S{i % len(_DOC_TOPICS) + 1}.{i:02d}",
            "fraud_categories": None,
            "policy_keywords": [w.lower() for w in title.split()],
            "char_count": len(chunk),
        })
    return out


_FEMA_SECTIONS = [
    ("Eligibility", "Assistance is limited to the applicant's primary "
     "residence; secondary homes are ineligible for housing assistance.",
     ["ineligible", "residence"]),
    ("Duplication of Benefits", "Assistance may not duplicate insurance "
     "payouts; insured losses must first be claimed against the policy.",
     ["duplicate", "insurance"]),
    ("Documentation", "Claims require an inspection or verifiable damage "
     "documentation; undocumented claims require additional records.",
     ["documentation"]),
    ("Fraud Indicators", "Shared bank accounts or phone numbers across "
     "multiple claims, and amounts far exceeding assessed damage, are "
     "fraud indicators subject to denial and referral.",
     ["fraud", "shared", "inflated"]),
    ("Award Limits", "Awards are capped at the assessed damage less any "
     "insurance amount; partial approvals apply when coverage overlaps.",
     ["partial", "limits"]),
]


def lab4_policy_docs(seed: int = 42, n_chunks: int = 40) -> list[dict]:
    """FEMA-policy-style chunks with the lab4 metadata schema
    (fraud_categories / policy_keywords — lab4 main.tf:270-290)."""
    rng = random.Random(seed)
    out = []
    for i in range(n_chunks):
        title, base, cats = _FEMA_SECTIONS[i % len(_FEMA_SECTIONS)]
        chunk = (f"{base} (guidance {i // len(_FEMA_SECTIONS) + 1}, ref "
                 + "".join(rng.choice("0123456789") for _ in range(6)) + ")")
        out.append({
            "document_id": f"POLICY-{i:04d}",
            "title": f"FEMA IHP {title}",
            "chunk": chunk,
            "pages": f"{10 + i}-{11 + i}",
            "section_reference": f"IHP-{i % len(_FEMA_SECTIONS) + 1}.{i:02d}",
            "fraud_categories": cats,
            "policy_keywords": [w.lower() for w in title.split()],
            "char_count": len(chunk),
        })
    return out


def publish_lab2(broker: Broker, seed: int = 42, n_chunks: int = 64) -> None:
    broker.create_topic("documents").purge()
    pd = AvroProducer(broker, "documents", schemas.DOCUMENTS)
    for d in lab2_documents(seed, n_chunks):
        pd.produce(d, key=d["document_id"], timestamp_ms=0, partition=0)
    broker.create_topic("queries")
    pq = AvroProducer(broker, "queries", schemas.QUERIES)
    pq.produce({"query": "How do I create a Flink table?"}, timestamp_ms=0,
               partition=0)
