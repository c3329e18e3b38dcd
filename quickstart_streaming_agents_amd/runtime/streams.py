"""HIP-stream operator pipelining (SURVEY.md 2.5 "pipeline parallelism
across operators" row).

The reference chains Flink jobs through Kafka topics; stages overlap
because they are separate cluster jobs.  Here chained CTAS statements
compose through in-process topics (sql/stream.py), and the GPU stages of
one pipeline — embed batch, index search, window/anomaly kernels —
overlap through HIP streams: each stage runs on its own
``torch.cuda.Stream``, batches flow stage-to-stage through CUDA events,
so stage i processes batch b while stage i+1 processes batch b-1
(classic software pipelining; with S stages and B >> S batches the wall
clock approaches max-stage instead of sum-of-stages).

CPU fallback runs the stages sequentially with identical results, so the
pipeline is testable without a device.
"""

from __future__ import annotations

from typing import Any, Callable, Sequence

import torch


class StreamPipeline:
    """Run `stages` (callables batch -> batch) over a batch iterator with
    one HIP stream per stage.

    Stage callables must do their GPU work on the CURRENT stream (true
    for torch ops and the qsa kernels, which launch on
    ``getCurrentHIPStream``) and return the value passed to the next
    stage.  Host-side stages are fine — they simply don't overlap.
    """

    def __init__(self, stages: Sequence[Callable[[Any], Any]]):
        assert stages, "need at least one stage"
        self.stages = list(stages)
        self.use_streams = torch.cuda.is_available()
        if self.use_streams:
            self.streams = [torch.cuda.Stream() for _ in self.stages]

    def run(self, batches: Sequence[Any]) -> list[Any]:
        if not self.use_streams:
            out = []
            for b in batches:
                for fn in self.stages:
                    b = fn(b)
                out.append(b)
            return out

        S = len(self.stages)

        def _record(v, stream) -> None:
            # a tensor crossing streams must be pinned to the consuming
            # stream, or the caching allocator may recycle its memory
            # once the PRODUCING stream's work retires
            if torch.is_tensor(v) and v.is_cuda:
                v.record_stream(stream)
            elif isinstance(v, (list, tuple)):
                for x in v:
                    _record(x, stream)

        # in_flight[s] = (value, event) waiting to enter stage s+1
        results: list[Any] = []
        # software pipeline: advance the deepest stages first each tick
        # so batch b's stage s runs concurrently with batch b+1's s-1
        slots: list[tuple[Any, torch.cuda.Event] | None] = [None] * S
        bi = 0
        n = len(batches)
        done = 0
        while done < n:
            for s in reversed(range(S)):
                if s == 0:
                    if bi < n and slots[0] is None:
                        with torch.cuda.stream(self.streams[0]):
                            v = self.stages[0](batches[bi])
                            ev = torch.cuda.Event()
                            ev.record(self.streams[0])
                        slots[0] = (v, ev)
                        bi += 1
                    continue
                if slots[s - 1] is not None and slots[s] is None:
                    v, ev = slots[s - 1]
                    slots[s - 1] = None
                    with torch.cuda.stream(self.streams[s]):
                        self.streams[s].wait_event(ev)
                        _record(v, self.streams[s])
                        v2 = self.stages[s](v)
                        ev2 = torch.cuda.Event()
                        ev2.record(self.streams[s])
                    slots[s] = (v2, ev2)
            # drain the last stage
            if slots[S - 1] is not None:
                v, ev = slots[S - 1]
                slots[S - 1] = None
                ev.synchronize()
                _record(v, torch.cuda.current_stream())
                results.append(v)
                done += 1
        return results


def pipelined_embed_index(encoder, index, chunks: list[dict],
                          batch_size: int = 256) -> int:
    """Build a vector index from document chunks with the embed and
    index-add stages overlapped on separate HIP streams (the lab2/lab4
    `documents -> embeddings -> index` ingestion path).  Returns the
    number of chunks added."""
    batches = [chunks[i:i + batch_size]
               for i in range(0, len(chunks), batch_size)]

    def embed(batch):
        vecs = encoder.embed_batch([c["chunk"] for c in batch])
        return batch, vecs

    def add(arg):
        batch, vecs = arg
        for c, v in zip(batch, vecs):
            meta = {k: val for k, val in c.items()
                    if k not in ("chunk", "embedding")}
            index.add(c.get("document_id", c.get("doc_id", "")),
                      c["chunk"], v, meta)
        return len(batch)

    pipe = StreamPipeline([embed, add])
    return sum(pipe.run(batches))
