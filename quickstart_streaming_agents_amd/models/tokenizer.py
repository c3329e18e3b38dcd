"""Deterministic air-gapped tokenizer (BASELINE.json: synthetic data /
random-init weights contract — token identity only has to be
deterministic, not linguistic).

No network means no real BPE vocab files; the north-star benchmark runs
random-init weights, so token IDENTITY only needs to be deterministic and
well-distributed, with realistic sequence lengths.  Words (and punctuation)
hash into the vocab above the reserved specials; token counts track
whitespace/punct splits, so prompt token lengths are realistic.
"""

from __future__ import annotations

import hashlib
import re
from functools import lru_cache

_SPLIT = re.compile(r"[A-Za-z0-9_$]+|[^\sA-Za-z0-9_]")


class HashTokenizer:
    PAD, BOS, EOS = 0, 1, 2
    N_SPECIAL = 16

    def __init__(self, vocab_size: int = 128_256):
        self.vocab_size = vocab_size

    @lru_cache(maxsize=65536)
    def _tok(self, piece: str) -> int:
        h = hashlib.blake2b(piece.encode(), digest_size=4).digest()
        return self.N_SPECIAL + int.from_bytes(h, "little") % (
            self.vocab_size - self.N_SPECIAL)

    def encode(self, text: str, bos: bool = True) -> list[int]:
        ids = [self.BOS] if bos else []
        ids.extend(self._tok(p) for p in _SPLIT.findall(text or ""))
        return ids

    def decode(self, ids: list[int]) -> str:
        # hash tokenization is lossy; decode to a stable placeholder stream
        return " ".join(f"<t{i}>" for i in ids)
