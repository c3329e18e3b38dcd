"""Continuous-batching serving engine for the agent LLM.

Sequences are admitted into the running decode batch as KV pages free up
(vLLM-style): each step() admits pending prompts (per-sequence prefill via
the GEMM+causal-softmax path), then runs ONE fused decode step for every
running sequence through the paged-attention kernel.  Thousands of agent
episodes (agents/schedule.py) gang their LLM turns into these batches; tool
round trips run on host threads between turns, so the GPU stays busy.

Greedy sampling keeps the benchmark deterministic.
"""

from __future__ import annotations

import time
from dataclasses import dataclass, field

import torch

from .kv_cache import PagedKVCache
from .llama import LlamaModel
from .tokenizer import HashTokenizer


@dataclass
class Sequence:
    seq_id: int
    prompt: list[int]
    max_new_tokens: int
    out_tokens: list[int] = field(default_factory=list)
    done: bool = False


@dataclass
class EngineStats:
    prefill_tokens: int = 0
    decode_tokens: int = 0
    decode_steps: int = 0
    prefills: int = 0
    wall_s: float = 0.0


class Engine:
    def __init__(self, model: LlamaModel, kv_pages: int | None = None,
                 max_batch: int = 256, max_seq_len: int = 4096,
                 eos_id: int | None = None):
        self.model = model
        self.max_batch = max_batch
        self.max_seq_len = max_seq_len
        self.eos_id = eos_id  # None -> run to max_new_tokens (random weights)
        if kv_pages is None:
            kv_pages = max_batch * ((max_seq_len + 63) // 64) + 64
        self.kv = model.new_kv_cache(kv_pages)
        self._next_id = 0
        self.pending: list[Sequence] = []
        self.running: list[Sequence] = []
        self.stats = EngineStats()

    # ------------------------------------------------------------------
    def submit(self, prompt_tokens: list[int], max_new_tokens: int) -> Sequence:
        seq = Sequence(self._next_id, list(prompt_tokens[: self.max_seq_len - 1]),
                       max_new_tokens)
        self._next_id += 1
        self.pending.append(seq)
        return seq

    def _admit(self) -> None:
        while self.pending and len(self.running) < self.max_batch:
            seq = self.pending[0]
            need = self.kv.pages_for(len(seq.prompt) + seq.max_new_tokens)
            if need > self.kv.free_pages:
                break
            self.pending.pop(0)
            self.kv.allocate(seq.seq_id, len(seq.prompt))
            toks = torch.tensor(seq.prompt, dtype=torch.int64,
                                device=self.model.device)
            logits = self.model.forward_prefill(toks, self.kv, seq.seq_id)
            first = int(torch.argmax(logits).item())
            seq.out_tokens.append(first)
            self.stats.prefill_tokens += len(seq.prompt)
            self.stats.prefills += 1
            self.running.append(seq)
            self._maybe_finish(seq)

    def _maybe_finish(self, seq: Sequence) -> None:
        if len(seq.out_tokens) >= seq.max_new_tokens or (
                self.eos_id is not None and seq.out_tokens
                and seq.out_tokens[-1] == self.eos_id):
            seq.done = True

    def step(self) -> int:
        """Admit + one decode step; returns number of running sequences."""
        self._admit()
        batch = [s for s in self.running if not s.done]
        if not batch:
            self.running = [s for s in self.running if not s.done]
            return 0
        dev = self.model.device
        # the token decoded this step is the last sampled one
        tokens = torch.tensor([s.out_tokens[-1] for s in batch],
                              dtype=torch.int64, device=dev)
        positions = torch.tensor(
            [len(s.prompt) + len(s.out_tokens) - 1 for s in batch],
            dtype=torch.int32, device=dev)
        for s in batch:
            self.kv.extend(s.seq_id, len(s.prompt) + len(s.out_tokens))
        seq_ids = [s.seq_id for s in batch]
        bt = self.kv.block_table(seq_ids)
        sl = self.kv.seq_lens_tensor(seq_ids)
        logits = self.model.forward_decode(tokens, self.kv, bt, sl, positions)
        nxt = torch.argmax(logits, dim=-1).tolist()
        for s, tok in zip(batch, nxt):
            s.out_tokens.append(int(tok))
            self._maybe_finish(s)
        self.stats.decode_tokens += len(batch)
        self.stats.decode_steps += 1
        done = [s for s in self.running if s.done]
        for s in done:
            self.kv.free(s.seq_id)
        self.running = [s for s in self.running if not s.done]
        return len(self.running) + len(self.pending)

    def run_to_completion(self) -> None:
        t0 = time.perf_counter()
        while self.pending or self.running:
            self.step()
        if torch.cuda.is_available():
            torch.cuda.synchronize()
        self.stats.wall_s += time.perf_counter() - t0

    def generate_batch(self, prompts: list[list[int]],
                       max_new_tokens: list[int]) -> list[list[int]]:
        seqs = [self.submit(p, m) for p, m in zip(prompts, max_new_tokens)]
        self.run_to_completion()
        return [s.out_tokens for s in seqs]


class EngineLLM:
    """llm_batch adapter for the lab pipelines: text in, text out."""

    def __init__(self, engine: Engine, tokenizer: HashTokenizer | None = None):
        self.engine = engine
        self.tokenizer = tokenizer or HashTokenizer(
            engine.model.cfg.vocab_size)

    def __call__(self, prompts: list[str], max_new_tokens: list[int]) -> list[str]:
        enc = [self.tokenizer.encode(p) for p in prompts]
        outs = self.engine.generate_batch(enc, list(max_new_tokens))
        return [self.tokenizer.decode(o) for o in outs]
