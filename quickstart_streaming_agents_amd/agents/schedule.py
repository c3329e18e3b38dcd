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
