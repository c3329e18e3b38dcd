"""Deterministic air-gapped tokenizers.

`BpeTokenizer` (the default on the bench path) is a real byte-level BPE:
merges trained deterministically on the synthetic lab corpus by
tools/train_bpe.py and committed to data/bpe_vocab.json — lossless
round-trip, subword token statistics in the Llama-3 class for in-domain
text (the reference's prompts are real English through Llama-3 BPE,
LAB1-Walkthrough.md:195-256), and reserved special tokens the
grammar-constrained agent decode uses as its decision vocabulary
(models/grammar.py): <|finish|> plus <|tool_0..7|> slots.

`HashTokenizer` remains for tests that only need deterministic token
identity (BASELINE.json: random-init weights contract).
"""

from __future__ import annotations

import hashlib
import json
import os
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


# ---------------------------------------------------------------------------
# Byte-level BPE (trained in-repo: tools/train_bpe.py)
# ---------------------------------------------------------------------------

_PRETOK = re.compile(
    r"'s|'t|'re|'ve|'m|'ll|'d| ?[A-Za-z]+| ?[0-9]+| ?[^\sA-Za-z0-9]+"
    r"|\s+(?!\S)|\s+")

_VOCAB_PATH = os.path.join(os.path.dirname(os.path.abspath(__file__)),
                           "..", "data", "bpe_vocab.json")


class BpeTokenizer:
    """Byte-level BPE with reserved special tokens.

    Layout: specials 0..15 (pad/bos/eot/finish/tool slots), byte tokens
    16..271 (byte value + 16), merged tokens from 272 up.  Lossless:
    decode(encode(text)) == text for any str.
    """

    PAD, BOS, EOS = 0, 1, 2
    FINISH = 3
    TOOL_SLOT0 = 4
    N_TOOL_SLOTS = 8
    N_SPECIAL = 16
    _BYTE0 = 16

    def __init__(self, vocab_size: int = 128_256,
                 vocab_path: str | None = None):
        with open(vocab_path or _VOCAB_PATH) as fh:
            v = json.load(fh)
        self.specials: dict[str, int] = v["specials"]
        self._special_by_id = {i: s for s, i in self.specials.items()}
        # merges are over a symbol space where 0..255 are raw bytes and
        # merge i creates symbol 256+i; token id = symbol + _BYTE0 for
        # bytes, or _BYTE0 + 256 + i for merge i.  A PREFIX of the merge
        # list is itself a valid (coarser) BPE, so small model vocabs
        # (tiny test presets) just truncate the merges — still lossless.
        assert vocab_size >= self.N_SPECIAL + 256, "vocab too small for BPE"
        merges = [tuple(p) for p in v["merges"]]
        merges = merges[: vocab_size - self.N_SPECIAL - 256]
        self._ranks: dict[tuple[int, int], int] = {
            p: i for i, p in enumerate(merges)}
        self._merge_sym = {p: 256 + i for i, p in enumerate(merges)}
        self.vocab_size = vocab_size
        self.n_tokens = self.N_SPECIAL + 256 + len(self._ranks)
        # symbol -> bytes, for decode
        self._sym_bytes: list[bytes] = [bytes([b]) for b in range(256)]
        for (a, b) in merges:
            self._sym_bytes.append(self._sym_bytes[a] + self._sym_bytes[b])
        self._special_re = re.compile(
            "(" + "|".join(re.escape(s) for s in self.specials) + ")")

    @lru_cache(maxsize=65536)
    def _bpe_word(self, word: str) -> tuple[int, ...]:
        syms = list(word.encode("utf-8"))
        while len(syms) > 1:
            best_rank, best_i = None, -1
            for i in range(len(syms) - 1):
                r = self._ranks.get((syms[i], syms[i + 1]))
                if r is not None and (best_rank is None or r < best_rank):
                    best_rank, best_i = r, i
            if best_rank is None:
                break
            pair = (syms[best_i], syms[best_i + 1])
            syms[best_i:best_i + 2] = [self._merge_sym[pair]]
        return tuple(s + self._BYTE0 for s in syms)

    def encode(self, text: str, bos: bool = True) -> list[int]:
        ids = [self.BOS] if bos else []
        for part in self._special_re.split(text or ""):
            if not part:
                continue
            sid = self.specials.get(part)
            if sid is not None:
                ids.append(sid)
                continue
            for word in _PRETOK.findall(part):
                ids.extend(self._bpe_word(word))
        return ids

    def decode(self, ids: list[int]) -> str:
        out: list[bytes] = []
        for i in ids:
            i = int(i)
            if i < self.N_SPECIAL:
                if i in (self.PAD, self.BOS):
                    continue
                out.append(self._special_by_id.get(
                    i, f"<|special_{i}|>").encode())
            elif i - self._BYTE0 < len(self._sym_bytes):
                out.append(self._sym_bytes[i - self._BYTE0])
            # ids beyond the trained vocab (random-weight sampling over the
            # full model vocab can produce them) decode to nothing
        return b"".join(out).decode("utf-8", errors="replace")

    def tool_slot(self, k: int) -> int:
        assert 0 <= k < self.N_TOOL_SLOTS
        return self.TOOL_SLOT0 + k


_DEFAULT: BpeTokenizer | None = None


def default_tokenizer(vocab_size: int = 128_256) -> BpeTokenizer:
    """Process-wide shared BPE instance (merge tables are read-only)."""
    global _DEFAULT
    if _DEFAULT is None or _DEFAULT.vocab_size != vocab_size:
        _DEFAULT = BpeTokenizer(vocab_size)
    return _DEFAULT
