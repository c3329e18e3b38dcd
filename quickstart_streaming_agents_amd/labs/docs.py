"""Knowledge-base document publishing (scripts/publish_docs.py parity).

The reference publishes markdown chunks with YAML frontmatter
(title / pages / section_reference / fraud_categories / policy_keywords /
char_count — publish_docs.py:63-109) onto a `documents` topic that the
vector store ingests.  This module chunks markdown files the same way:
`---` frontmatter blocks parsed (flat YAML subset, no dependency), body
split into chunks, records produced in the documents schema.
"""

from __future__ import annotations

import os
import re

from ..wire import AvroProducer, Broker
from . import schemas


def parse_frontmatter(text: str) -> tuple[dict, str]:
    """Parse a leading `---` YAML frontmatter block (flat keys, scalar or
    [a, b] list values) -> (meta, body)."""
    m = re.match(r"\A---\s*\n(.*?)\n---\s*\n?(.*)\Z", text, re.DOTALL)
    if not m:
        return {}, text
    meta: dict = {}
    for line in m.group(1).splitlines():
        line = line.strip()
        if not line or line.startswith("#") or ":" not in line:
            continue
        key, _, val = line.partition(":")
        val = val.strip()
        if val.startswith("[") and val.endswith("]"):
            items = [v.strip().strip("'\"") for v in val[1:-1].split(",")]
            meta[key.strip()] = [v for v in items if v]
        else:
            meta[key.strip()] = val.strip("'\"")
    return meta, m.group(2)


def chunk_markdown(text: str, max_chars: int = 1500) -> list[str]:
    """Split a markdown body into chunks on headings, packing adjacent
    sections up to max_chars (publish_docs chunking shape)."""
    sections = re.split(r"(?m)^(?=#{1,3} )", text)
    chunks: list[str] = []
    cur = ""
    for sec in sections:
        sec = sec.strip()
        if not sec:
            continue
        if cur and len(cur) + len(sec) + 1 > max_chars:
            chunks.append(cur)
            cur = sec
        else:
            cur = f"{cur}\n{sec}" if cur else sec
        while len(cur) > max_chars:
            chunks.append(cur[:max_chars])
            cur = cur[max_chars:]
    if cur:
        chunks.append(cur)
    return chunks


def doc_records(path: str, max_chars: int = 1500) -> list[dict]:
    """One markdown file -> documents-topic records."""
    with open(path) as fh:
        meta, body = parse_frontmatter(fh.read())
    stem = os.path.splitext(os.path.basename(path))[0]
    out = []
    for i, chunk in enumerate(chunk_markdown(body, max_chars)):
        out.append({
            "document_id": f"{stem}-{i:04d}",
            "chunk": chunk,
            "title": meta.get("title", stem),
            "pages": str(meta.get("pages", "")),
            "section_reference": meta.get("section_reference"),
            "fraud_categories": meta.get("fraud_categories", []) or [],
            "policy_keywords": meta.get("policy_keywords", []) or [],
            "char_count": len(chunk),
        })
    return out


def publish_docs(broker: Broker, paths: list[str],
                 topic: str = "documents", max_chars: int = 1500,
                 clear_first: bool = True) -> int:
    """Publish markdown files as Avro document chunks
    (publish_docs.py flow: optional clear, then produce)."""
    if clear_first and topic in broker.topics:
        broker.topic(topic).purge()
    producer = AvroProducer(broker, topic, schemas.DOCUMENTS)
    n = 0
    for p in sorted(paths):
        for rec in doc_records(p, max_chars):
            producer.produce(rec, key=rec["document_id"], partition=0)
            n += 1
    return n
