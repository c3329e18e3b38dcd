"""VECTOR_SEARCH_AGG semantics on the CPU reference path (the numerics
reference for ops/hip/topk_cosine.hip): exact cosine top-k, 1536-d
embedding contract, MongoDB (1+cos)/2 score parity, cache invalidation
on add (lab2 main.tf:215,292 per SURVEY.md 2.4 K2)."""

import numpy as np

from quickstart_streaming_agents_amd.vector.index import (
    EMBED_DIM, HashingEmbedder, VectorIndex, vector_search_agg)


def _idx(n=20, dim=8, seed=0):
    rng = np.random.default_rng(seed)
    idx = VectorIndex(dim=dim)
    for i in range(n):
        idx.add(f"d{i}", f"chunk {i}", rng.standard_normal(dim),
                {"pages": [i]})
    return idx


def test_embedding_dims_contract():
    assert EMBED_DIM == 1536
    v = HashingEmbedder().embed("hello")
    assert v.shape == (1536,) and v.dtype == np.float32
    # deterministic and unit-normalized
    assert np.allclose(v, HashingEmbedder().embed("hello"))
    assert abs(float(np.linalg.norm(v)) - 1.0) < 1e-5


def test_self_hit_is_top1_with_cosine_1():
    idx = _idx()
    q = idx.matrix[7] * 5.0          # un-normalized query, same direction
    hits = idx.search(q, k=3)
    assert hits[0].document_id == "d7"
    assert abs(hits[0].score - 1.0) < 1e-5
    assert hits[0].metadata == {"pages": [7]}
    # scores are sorted descending
    assert hits[0].score >= hits[1].score >= hits[2].score


def test_k_larger_than_index_and_empty():
    idx = _idx(n=2)
    assert len(idx.search(np.ones(8), k=10)) == 2
    empty = VectorIndex(dim=8)
    assert empty.search(np.ones(8), k=3) == []
    assert empty.search_batch(np.ones((4, 8)), k=3) == [[], [], [], []]


def test_add_invalidates_matrix_cache():
    idx = _idx(n=3)
    assert idx.matrix.shape == (3, 8)
    idx.add("new", "late row", np.ones(8))
    assert idx.matrix.shape == (4, 8)
    assert idx.search(np.ones(8), k=1)[0].document_id == "new"


def test_mongo_score_parity():
    assert VectorIndex.mongo_score(1.0) == 1.0
    assert VectorIndex.mongo_score(-1.0) == 0.0
    assert VectorIndex.mongo_score(0.0) == 0.5


def test_vector_search_agg_matches_search():
    idx = _idx()
    q = np.random.default_rng(1).standard_normal(8)
    assert [h.document_id for h in vector_search_agg(idx, q, 3)] == \
        [h.document_id for h in idx.search(q, 3)]


def test_batch_matches_single():
    idx = _idx(n=50)
    qs = np.random.default_rng(2).standard_normal((5, 8))
    batch = idx.search_batch(qs, k=4)
    for q, hits in zip(qs, batch):
        assert [h.document_id for h in hits] == \
            [h.document_id for h in idx.search(q, 4)]


def test_add_documents_embeds_missing_and_keeps_metadata():
    emb = HashingEmbedder()
    idx = VectorIndex()
    idx.add_documents([
        {"document_id": "a", "chunk": "flink windows",
         "title": "Windows", "pages": [1, 2]},
        {"document_id": "b", "chunk": "flink joins",
         "embedding": emb.embed("flink joins"), "title": "Joins"},
    ], emb)
    hits = idx.search(emb.embed("flink joins"), k=1)
    assert hits[0].document_id == "b"
    assert hits[0].metadata["title"] == "Joins"
