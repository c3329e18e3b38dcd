"""Vector index + VECTOR_SEARCH_AGG semantics.

The reference delegates vector search to MongoDB Atlas `$vectorSearch`
(1536-dim cosine, top-k=3, numCandidates=500, index must be READY with
path=embedding — scripts/common/validate.py:56-210; lab2 main.tf:215,292).
Here the index is resident in GPU HBM (torch bf16/f32 matrix, L2-normalized
rows) with exact cosine top-k computed by the hand-written HIP kernel
(ops/hip/topk_cosine.hip); this module provides the index container, the
CPU/numpy exact reference, and the deterministic hashing embedder used for
air-gapped CPU runs (the GPU embedding path is models/encoder.py).

Sharding: with DP>1 the row space shards across ranks; cross-shard top-k
merges local (score, id) heaps via all-gather (parallel/shard_index.py).
"""

from __future__ import annotations

import hashlib
import re
from dataclasses import dataclass, field

import numpy as np

EMBED_DIM = 1536  # the reference's validated contract (validate.py:56-62)


@dataclass
class SearchHit:
    document_id: str
    chunk: str
    score: float
    metadata: dict = field(default_factory=dict)


class HashingEmbedder:
    """Deterministic 1536-d text embedder (feature-hashed word n-grams).

    The air-gapped CPU stand-in for the managed embedding models
    (titan-embed-v1 / ada-002, both 1536-d): same text -> same vector,
    shared tokens -> high cosine.  The GPU path (models/encoder.py) is a
    real transformer encoder; both produce L2-normalized EMBED_DIM vectors.
    """

    def __init__(self, dim: int = EMBED_DIM):
        self.dim = dim

    def embed(self, text: str) -> np.ndarray:
        vec = np.zeros(self.dim, dtype=np.float32)
        words = re.findall(r"[a-z0-9_]+", (text or "").lower())
        grams = words + [f"{a}_{b}" for a, b in zip(words, words[1:])]
        for g in grams:
            h = hashlib.blake2b(g.encode(), digest_size=8).digest()
            idx = int.from_bytes(h[:4], "little") % self.dim
            sign = 1.0 if h[4] & 1 else -1.0
            vec[idx] += sign
        n = float(np.linalg.norm(vec))
        if n > 0:
            vec /= n
        return vec

    def embed_batch(self, texts: list[str]) -> np.ndarray:
        return np.stack([self.embed(t) for t in texts]) if texts else \
            np.zeros((0, self.dim), dtype=np.float32)


class VectorIndex:
    """Exact cosine top-k over L2-normalized rows.

    CPU store is numpy; `to_torch(device)` materializes the HBM-resident
    copy the HIP kernel searches.  Scores are cosine similarity in [-1, 1]
    (MongoDB's vectorSearch cosine score is (1+cos)/2; we keep raw cosine
    and expose `mongo_score` for parity assertions).
    """

    def __init__(self, dim: int = EMBED_DIM):
        self.dim = dim
        self._vecs: list[np.ndarray] = []
        self.ids: list[str] = []
        self.chunks: list[str] = []
        self.metadata: list[dict] = []
        self._matrix: np.ndarray | None = None
        self._torch_matrix = None
        self._torch_device = None

    def __len__(self) -> int:
        return len(self.ids)

    def add(self, doc_id: str, chunk: str, embedding: np.ndarray,
            metadata: dict | None = None) -> None:
        v = np.asarray(embedding, dtype=np.float32)
        n = float(np.linalg.norm(v))
        if n > 0:
            v = v / n
        self._vecs.append(v)
        self.ids.append(doc_id)
        self.chunks.append(chunk)
        self.metadata.append(metadata or {})
        self._matrix = None
        self._torch_matrix = None

    def add_documents(self, docs: list[dict], embedder) -> None:
        for d in docs:
            emb = d.get("embedding")
            if emb is None:
                emb = embedder.embed(d["chunk"])
            meta = {k: v for k, v in d.items()
                    if k not in ("document_id", "chunk", "embedding")}
            self.add(d["document_id"], d["chunk"], np.asarray(emb), meta)

    @property
    def matrix(self) -> np.ndarray:
        if self._matrix is None:
            self._matrix = (np.stack(self._vecs) if self._vecs
                            else np.zeros((0, self.dim), dtype=np.float32))
        return self._matrix

    # ---- CPU exact search (numerics reference for the HIP kernel) --------
    def search(self, query: np.ndarray, k: int = 3) -> list[SearchHit]:
        if not self._vecs:
            return []
        if self._torch_matrix is not None and \
                str(self._torch_device).startswith("cuda"):
            return self.search_batch_gpu(np.asarray(query)[None, :], k,
                                         device=self._torch_device)[0]
        q = np.asarray(query, dtype=np.float32)
        n = float(np.linalg.norm(q))
        if n > 0:
            q = q / n
        scores = self.matrix @ q
        k = min(k, len(scores))
        top = np.argpartition(-scores, k - 1)[:k]
        top = top[np.argsort(-scores[top], kind="stable")]
        return [SearchHit(self.ids[i], self.chunks[i], float(scores[i]),
                          self.metadata[i]) for i in top]

    def search_batch(self, queries: np.ndarray, k: int = 3) -> list[list[SearchHit]]:
        if self._torch_matrix is not None and \
                str(self._torch_device).startswith("cuda"):
            return self.search_batch_gpu(np.asarray(queries), k,
                                         device=self._torch_device)
        return [self.search(q, k) for q in np.asarray(queries)]

    # ---- GPU path --------------------------------------------------------
    def to_torch(self, device: str = "cuda", dtype=None):
        import torch
        if self._torch_matrix is None or self._torch_device != device:
            dtype = dtype or torch.float32
            self._torch_matrix = torch.from_numpy(self.matrix.copy()).to(
                device=device, dtype=dtype)
            self._torch_device = device
        return self._torch_matrix

    def search_batch_gpu(self, queries: np.ndarray, k: int = 3,
                         device: str = "cuda") -> list[list[SearchHit]]:
        """Exact cosine top-k on the HBM-resident matrix via the HIP
        tiled-dot-product/top-k kernel (ops/hip/topk_cosine.hip)."""
        import torch

        from ..ops import dispatch as D
        if not len(self):
            return [[] for _ in range(len(queries))]
        q = np.asarray(queries, dtype=np.float32)
        norms = np.linalg.norm(q, axis=1, keepdims=True)
        q = q / np.maximum(norms, 1e-12)
        docs = self.to_torch(device)
        qt = torch.from_numpy(q).to(device=device, dtype=torch.float32)
        k_eff = min(k, len(self))
        if len(q) >= 16:
            # large query batches are GEMM-shaped: one MFMA matmul over
            # the HBM matrix + top-k beats per-query matrix re-streaming
            scores_all = qt @ docs.T
            scores, idx = torch.topk(scores_all, k_eff, dim=1)
            idx = idx.int()
        else:
            # small/latency path: the tiled cosine/top-k HIP kernel
            scores, idx = D.topk_cosine(qt.contiguous(), docs, k_eff)
        scores = scores.cpu().numpy()
        idx = idx.cpu().numpy()
        out = []
        for qi in range(len(q)):
            out.append([SearchHit(self.ids[i], self.chunks[i],
                                  float(scores[qi, j]), self.metadata[i])
                        for j, i in enumerate(idx[qi])])
        return out

    @staticmethod
    def mongo_score(cosine: float) -> float:
        return (1.0 + cosine) / 2.0


def vector_search_agg(index: VectorIndex, embedding: np.ndarray,
                      k: int = 3) -> list[SearchHit]:
    """VECTOR_SEARCH_AGG(table, DESCRIPTOR(embedding), vec, k): top-k hits
    with scores + metadata columns unpacked downstream (lab2 main.tf:292)."""
    return index.search(embedding, k)
