"""Topic capture -> file, and file -> topic replay with timestamp rebase.

Reference parity: scripts/capture_lab1_data.py / capture_lab3_data.py
(consume topics, decode the Confluent wire format, write files for
distribution) and scripts/publish_lab3_data.py:143-170 /
lab4_datagen.py:45-59 (replay captured files with timestamps REBASED so
the data spans an exact number of aligned windows ending just past "now"
— the determinism trick that makes window 288 close and the anomaly fire).
"""

from __future__ import annotations

import base64
import json
import time

from ..wire import AvroConsumer, AvroProducer, Broker


def capture_topic(broker: Broker, topic: str, schema, path: str) -> int:
    """Consume a topic and write JSONL: base64 raw payload + decoded value
    + timestamp (capture_lab3_data.py shape)."""
    consumer = AvroConsumer(broker, topic, schema)
    n = 0
    with open(path, "w") as fh:
        for rec, value in consumer.poll():
            raw = rec.value if isinstance(rec.value, (bytes, bytearray)) \
                else json.dumps(value).encode()
            fh.write(json.dumps({
                "key": rec.key if not isinstance(rec.key, (bytes, bytearray))
                else base64.b64encode(rec.key).decode(),
                "value_b64": base64.b64encode(bytes(raw)).decode(),
                "value": value,
                "timestamp_ms": rec.timestamp_ms,
                "partition": rec.partition,
            }) + "\n")
            n += 1
    return n


def compute_rebase_offset(timestamps_ms: list[int], window_ms: int,
                          n_windows: int | None = None,
                          now_ms: int | None = None,
                          slack_ms: int = 10_000) -> int:
    """Offset so the data ends `slack_ms` past an aligned window boundary.

    publish_lab3_data.py:143-170 semantics: align so the records span
    exactly N x window_ms windows whose LAST window closes (its end +
    slack is in the past relative to the rebased "now"), making the final
    (spike) window emit deterministically.
    """
    if not timestamps_ms:
        return 0
    now_ms = int(time.time() * 1000) if now_ms is None else now_ms
    t_max = max(timestamps_ms)
    if n_windows is None:
        t_min = min(timestamps_ms)
        n_windows = max(1, (t_max - t_min) // window_ms + 1)
    # target end: the window boundary at/just before (now - slack), plus
    # slack -> data ends slack past an aligned boundary
    boundary = ((now_ms - slack_ms) // window_ms) * window_ms
    target_end = boundary + slack_ms
    return target_end - t_max


def replay_file(broker: Broker, path: str, topic: str, schema,
                ts_field: str, window_ms: int | None = None,
                now_ms: int | None = None, purge: bool = True) -> int:
    """Replay a captured JSONL file onto a topic: purge-then-publish,
    chronological sort, optional window-aligned timestamp rebase
    (the lab3/lab4 recovery/determinism pattern)."""
    rows = []
    with open(path) as fh:
        for line in fh:
            line = line.strip()
            if line:
                rows.append(json.loads(line))
    values = [r["value"] for r in rows]
    ts = [int(v[ts_field]) for v in values]
    offset = 0
    if window_ms:
        offset = compute_rebase_offset(ts, window_ms, now_ms=now_ms)
    order = sorted(range(len(values)), key=lambda i: ts[i])
    if purge and topic in broker.topics:
        broker.topic(topic).purge()
    producer = AvroProducer(broker, topic, schema)
    for i in order:
        v = dict(values[i])
        v[ts_field] = ts[i] + offset
        producer.produce(v, key=rows[i].get("key"),
                         timestamp_ms=v[ts_field], partition=0)
    return len(order)
