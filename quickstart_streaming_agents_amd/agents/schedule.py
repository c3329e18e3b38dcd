"""Episode scheduler: batch LLM requests across concurrent agent episodes.

The reference's AI_RUN_AGENT runs one managed-LLM round trip per iteration
per record.  Here thousands of per-record episodes run
concurrently and their ("llm", prompt) requests are ganged into batched
model calls while ("tool", ...) requests run on an I/O thread pool — the
model (and later the GPU decode engine, models/serve.py) always sees large
batches, and episodes blocked on tool I/O never stall the batch.
"""

from __future__ import annotations

from concurrent.futures import ThreadPoolExecutor
from typing import Any, Callable, Generator

from .runner import EpisodeResult


def run_episodes(episodes: list[Generator],
                 llm_batch: Callable[[list[str], list[int]], list[str]],
                 tool: Callable[[str, dict], str],
                 max_tool_workers: int = 16) -> list[EpisodeResult]:
    """Drive episodes to completion with ganged LLM batches.

    llm_batch(prompts, max_new_tokens_list) -> list of generated texts.
    tool(name, args) -> result text (exceptions become __error__ results).
    """
    import time as _time
    results: dict[int, EpisodeResult] = {}
    # (episode index -> pending request), advanced in rounds
    pending: dict[int, tuple] = {}
    started: dict[int, float] = {}

    def _advance(idx: int, send_value: Any) -> None:
        try:
            req = episodes[idx].send(send_value)
            pending[idx] = req
        except StopIteration as stop:
            results[idx] = stop.value
            if stop.value is not None and idx in started:
                # per-decision end-to-end latency (submit -> finish)
                stop.value.latency_s = _time.perf_counter() - started[idx]
            pending.pop(idx, None)
            if hasattr(llm_batch, "release"):
                llm_batch.release(idx)  # free the conversation's KV prefix

    for idx in range(len(episodes)):
        started[idx] = _time.perf_counter()
        _advance(idx, None)

    # conversation-aware backends (EngineLLM) take conv ids for KV prefix
    # reuse across an episode's turns; plain callables take 2 args.
    conv_aware = hasattr(llm_batch, "release")

    with ThreadPoolExecutor(max_workers=max_tool_workers) as pool:
        while pending:
            llm_ids = [i for i, r in pending.items() if r[0] == "llm"]
            tool_ids = [i for i, r in pending.items() if r[0] == "tool"]
            # Tools first (their futures overlap the LLM batch below).
            tool_futs = {}
            for i in tool_ids:
                _, name, args = pending[i]
                tool_futs[i] = pool.submit(_safe_tool, tool, name, args)
            if llm_ids:
                prompts = [pending[i][1] for i in llm_ids]
                maxtoks = [pending[i][2] for i in llm_ids]
                grammars = [pending[i][3] if len(pending[i]) > 3 else None
                            for i in llm_ids]
                if conv_aware:
                    if any(g is not None for g in grammars):
                        texts = llm_batch(prompts, maxtoks, llm_ids,
                                          grammars)
                    else:
                        texts = llm_batch(prompts, maxtoks, llm_ids)
                else:
                    texts = llm_batch(prompts, maxtoks)
                for i, text in zip(llm_ids, texts):
                    _advance(i, text)
            for i, fut in tool_futs.items():
                _advance(i, fut.result())
    if hasattr(llm_batch, "release_all"):
        llm_batch.release_all()
    return [results[i] for i in range(len(episodes))]


def _safe_tool(tool: Callable[[str, dict], str], name: str, args: dict) -> str:
    try:
        return tool(name, args)
    except Exception as e:
        return f"__error__ {e}"


def run_episodes_continuous(episodes: list[Generator], llm,
                            tool: Callable[[str, dict], str],
                            max_tool_workers: int = 16,
                            decode_chunk: int = 8,
                            tracer=None) -> list[EpisodeResult]:
    """Event-driven episode scheduler over the continuous-batching engine.

    `run_episodes` advances episodes in lockstep ROUNDS: every episode's
    turn N completes before any turn N+1 starts, so late rounds decode
    tiny batches (the engine is continuous but the driver isn't).  This
    scheduler is truly continuous: episode turns are submitted the
    moment they are ready (episode start, tool-future completion), the
    engine runs in short admit+decode slices (Engine.run_chunk), and
    completed turns re-enter their episodes immediately — a finish turn
    of one episode decodes in the same batch as turn-2 prefills of
    another, and per-decision latency stops being quantized to rounds.

    Requires the conversation-aware EngineLLM (submit_turn/pop_finished).
    tracer: optional runtime.trace.Tracer — one span per episode
    (submit -> finish wall time, iterations/tool calls in meta).
    """
    import time as _time
    results: dict[int, EpisodeResult] = {}
    started: dict[int, float] = {}
    tool_futs: dict[int, object] = {}
    n = len(episodes)

    def _advance(idx: int, send_value: Any, pool) -> None:
        try:
            req = episodes[idx].send(send_value)
        except StopIteration as stop:
            results[idx] = stop.value
            if stop.value is not None:
                stop.value.latency_s = _time.perf_counter() - started[idx]
                if tracer is not None and tracer.enabled:
                    from ..runtime.trace import Span
                    tracer.spans.append(Span(
                        stage=f"episode[{idx}]", t0=started[idx],
                        dt=stop.value.latency_s, records_in=1,
                        records_out=1,
                        meta={"status": stop.value.status,
                              "iterations": stop.value.iterations,
                              "tool_calls": stop.value.tool_calls}))
            llm.release(idx)
            return
        if req[0] == "llm":
            grammar = req[3] if len(req) > 3 else None
            llm.submit_turn(idx, req[1], req[2], grammar)
        elif req[0] == "tool":
            tool_futs[idx] = pool.submit(_safe_tool, tool, req[1], req[2])
        else:
            raise ValueError(f"unknown request {req[0]!r}")

    eng = llm.engine
    with ThreadPoolExecutor(max_workers=max_tool_workers) as pool:
        for idx in range(n):
            started[idx] = _time.perf_counter()
            _advance(idx, None, pool)
        while len(results) < n:
            # tool completions first: their turns join this slice's admit
            done_futs = [i for i, f in tool_futs.items() if f.done()]
            for i in done_futs:
                f = tool_futs.pop(i)
                _advance(i, f.result(), pool)
            if eng.pending or eng.running:
                eng.run_chunk(decode_chunk)
                for cid, text in llm.pop_finished():
                    _advance(cid, text, pool)
            elif tool_futs:
                # nothing on the GPU: block briefly on tool I/O
                _time.sleep(0.0005)
    llm.release_all()
    return [results[i] for i in range(n)]
