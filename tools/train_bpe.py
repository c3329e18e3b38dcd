#!/usr/bin/env python3
"""Train the deterministic in-repo byte-level BPE vocabulary.

The reference's prompts are real English text tokenized by Llama-3's BPE
(LAB1-Walkthrough.md:195-256); the air-gapped framework cannot ship that
vocab, so it trains its own byte-level BPE on the synthetic lab corpus
(datagen texts, agent prompts, competitor HTML, tool-call JSON, docs) —
fully deterministic (fixed corpus, fixed tie-breaks) — and commits the
result to quickstart_streaming_agents_amd/data/bpe_vocab.json.

Run from the repo root:  python tools/train_bpe.py [--merges N]
"""

from __future__ import annotations

import argparse
import collections
import json
import os
import re
import sys

ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
sys.path.insert(0, ROOT)

# GPT-2-style pre-tokenization (contractions, words w/ leading space,
# numbers, punctuation runs, whitespace)
PRETOK = re.compile(
    r"'s|'t|'re|'ve|'m|'ll|'d| ?[A-Za-z]+| ?[0-9]+| ?[^\sA-Za-z0-9]+|\s+(?!\S)|\s+")


def build_corpus() -> str:
    from quickstart_streaming_agents_amd.agents.mcp import competitor_html
    from quickstart_streaming_agents_amd.labs import datagen, pipelines

    parts: list[str] = []
    # docs / README: generic technical English
    for p in ("README.md", "ARCHITECTURE.md", "docs/SQL.md",
              "docs/MODELS.md", "docs/OPERATIONS.md"):
        fp = os.path.join(ROOT, p)
        if os.path.exists(fp):
            with open(fp) as fh:
                parts.append(fh.read())
    # lab fixtures
    parts.append(competitor_html())
    prods = datagen.lab1_products()
    custs = datagen.lab1_customers()
    parts += [json.dumps(p) for p in prods]
    parts += [json.dumps(c) for c in custs[:10]]
    parts.append(pipelines.LAB1_AGENT_PROMPT)
    for o in datagen.lab1_orders():
        row = dict(o)
        row["product_name"] = prods[0]["product_name"]
        row["order_price"] = prods[0]["price"]
        parts.append(pipelines.lab1_user_prompt(
            row, "http://127.0.0.1:8000/competitor", "user@example.com"))
    for c in datagen.lab4_claims()[:300]:
        parts.append(c["claim_narrative"])
        parts.append(json.dumps(c))
    for d in datagen.lab2_documents():
        parts.append(d["chunk"])
    for d in datagen.lab4_policy_docs():
        parts.append(d["chunk"])
    # tool-call syntax the agent loop emits
    for name in ("http_get", "http_post", "send_email"):
        parts.append(
            'TOOL_CALL {"name": "%s", "arguments": {"url": '
            '"http://127.0.0.1:8000/competitor"}}' % name)
    parts.append('TOOL_CALL {"name": "send_email", "arguments": {"to": '
                 '"user@example.com", "subject": "Price Match Applied", '
                 '"body": "A refund for the difference is on its way."}}')
    parts.append("Competitor Price:\n209.99\n\nDecision:\nPRICE_MATCH\n\n"
                 "Summary:\nFound competitor price below ours; sent a "
                 "price match email.")
    return "\n".join(parts)


def train(corpus: str, n_merges: int) -> list[list[int]]:
    """Classic BPE over the word-frequency dict; token unit = byte value.
    Deterministic: ties break on (count desc, pair asc)."""
    words = collections.Counter(PRETOK.findall(corpus))
    # each word -> tuple of symbols (ints; merged symbols get new ids)
    seqs = {w: tuple(w.encode("utf-8")) for w in words}
    merges: list[list[int]] = []
    next_id = 256
    for _ in range(n_merges):
        pairs: collections.Counter = collections.Counter()
        for w, seq in seqs.items():
            f = words[w]
            for a, b in zip(seq, seq[1:]):
                pairs[(a, b)] += f
        if not pairs:
            break
        best = min(pairs.items(), key=lambda kv: (-kv[1], kv[0]))
        (a, b), cnt = best
        if cnt < 2:
            break
        merges.append([a, b])
        new = next_id
        next_id += 1
        for w, seq in list(seqs.items()):
            if a not in seq:
                continue
            out = []
            i = 0
            while i < len(seq):
                if i + 1 < len(seq) and seq[i] == a and seq[i + 1] == b:
                    out.append(new)
                    i += 2
                else:
                    out.append(seq[i])
                    i += 1
            seqs[w] = tuple(out)
    return merges


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--merges", type=int, default=8000)
    args = ap.parse_args()
    corpus = build_corpus()
    print(f"corpus: {len(corpus)} chars")
    merges = train(corpus, args.merges)
    print(f"trained {len(merges)} merges")
    out = {
        "version": 1,
        "n_special": 16,
        "specials": {
            "<|pad|>": 0, "<|bos|>": 1, "<|eot|>": 2, "<|finish|>": 3,
            "<|tool_0|>": 4, "<|tool_1|>": 5, "<|tool_2|>": 6,
            "<|tool_3|>": 7, "<|tool_4|>": 8, "<|tool_5|>": 9,
            "<|tool_6|>": 10, "<|tool_7|>": 11,
        },
        "merges": merges,
    }
    dst = os.path.join(ROOT, "quickstart_streaming_agents_amd", "data",
                       "bpe_vocab.json")
    with open(dst, "w") as fh:
        json.dump(out, fh)
    print(f"wrote {dst} ({os.path.getsize(dst)} bytes)")

    # quick stats on held-out-ish text
    from quickstart_streaming_agents_amd.models.tokenizer import BpeTokenizer
    tok = BpeTokenizer()
    sample = corpus[len(corpus) // 3: len(corpus) // 3 + 20000]
    ids = tok.encode(sample, bos=False)
    print(f"chars/token on corpus slice: {len(sample) / max(1, len(ids)):.2f}")
    rt = tok.decode(ids)
    assert rt == sample, "round-trip failed"
    print("round-trip OK")


if __name__ == "__main__":
    main()
