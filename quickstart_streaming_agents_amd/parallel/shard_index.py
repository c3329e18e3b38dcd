"""Cross-shard vector retrieval (K2 at DP>1).

The reference's VECTOR_SEARCH_AGG hits one managed MongoDB Atlas index
(lab2 main.tf:215,292 — top-k=3, cosine).  At DP>1 the HBM-resident index
shards row-wise across ranks (288 GB/GPU => each shard holds its slice of
the corpus); a query computes LOCAL exact top-k with the HIP cosine/top-k
kernel, then merges the tiny (score, hit) candidate sets across ranks.

k is 3 and a hit is <2 KB, so the exchange is latency- not bandwidth-bound:
one all_gather_object over xGMI per query batch, merge on host.  Every rank
returns the same global top-k (all ranks run the same downstream pipeline
for their own stream partitions).
"""

from __future__ import annotations

import numpy as np
import torch.distributed as dist

from ..vector.index import SearchHit, VectorIndex


class ShardedVectorIndex:
    """Row-sharded VectorIndex: doc i lives on rank i % world_size."""

    def __init__(self, world_size: int = 1, rank: int = 0, group=None,
                 dim: int | None = None):
        kw = {} if dim is None else {"dim": dim}
        self.local = VectorIndex(**kw)
        self.world_size = world_size
        self.rank = rank
        self.group = group
        self._added = 0

    def __len__(self) -> int:
        return self._added

    def add_documents(self, docs: list[dict], embedder) -> None:
        """Same call surface as VectorIndex; keeps only this rank's rows.
        Every rank must call with the SAME docs in the SAME order."""
        for i, d in enumerate(docs, start=self._added):
            if i % self.world_size == self.rank:
                emb = d.get("embedding")
                if emb is None:
                    emb = embedder.embed(d["chunk"])
                meta = {k: v for k, v in d.items()
                        if k not in ("document_id", "chunk", "embedding")}
                self.local.add(d["document_id"], d["chunk"],
                               np.asarray(emb), meta)
        self._added += len(docs)

    def search(self, query: np.ndarray, k: int = 3) -> list[SearchHit]:
        return self.search_batch(np.asarray(query)[None, :], k)[0]

    def search_batch(self, queries: np.ndarray,
                     k: int = 3) -> list[list[SearchHit]]:
        local = self.local.search_batch(np.asarray(queries), k)
        if self.world_size <= 1:
            return local
        gathered: list = [None] * self.world_size
        dist.all_gather_object(gathered, local, group=self.group)
        out: list[list[SearchHit]] = []
        for qi in range(len(queries)):
            cands: list[SearchHit] = []
            for rank_hits in gathered:
                cands.extend(rank_hits[qi])
            cands.sort(key=lambda h: (-h.score, h.document_id))
            out.append(cands[:k])
        return out
