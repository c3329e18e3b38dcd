"""HIP-stream operator pipelining (runtime/streams.py): correctness on
CPU (sequential fallback) and equivalence of the pipelined index build
with the sequential one."""

import pytest

from quickstart_streaming_agents_amd.runtime.streams import StreamPipeline


def test_stream_pipeline_cpu_sequential():
    pipe = StreamPipeline([lambda x: x + 1, lambda x: x * 2,
                           lambda x: x - 3])
    assert pipe.run(list(range(10))) == [(x + 1) * 2 - 3
                                         for x in range(10)]


def test_pipelined_index_build_matches_sequential():
    from quickstart_streaming_agents_amd.labs import datagen
    from quickstart_streaming_agents_amd.runtime.streams import \
        pipelined_embed_index
    from quickstart_streaming_agents_amd.vector.index import (HashingEmbedder,
                                                              VectorIndex)
    docs = [{"document_id": f"d{i}", "chunk": c["chunk"]}
            for i, c in enumerate(datagen.lab2_documents(n_chunks=32))]
    emb = HashingEmbedder()

    seq = VectorIndex()
    seq.add_documents([dict(d) for d in docs], emb)
    pip = VectorIndex()
    n = pipelined_embed_index(emb, pip, [dict(d) for d in docs],
                              batch_size=7)
    assert n == len(docs)
    assert pip.ids == seq.ids
    import numpy as np
    q = emb.embed_batch(["flink table windows"])[0]
    hs, hp = seq.search(q, 3), pip.search(q, 3)
    assert [h.document_id for h in hs] == [h.document_id for h in hp]
    assert np.allclose([h.score for h in hs], [h.score for h in hp])
