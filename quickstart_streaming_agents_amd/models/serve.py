"""Continuous-batching serving engine for the agent LLM — the local
replacement for the managed ML_PREDICT('llm_textgen_model') endpoint
(reference terraform/core/main.tf:461,495 models).

Sequences are admitted into the running decode batch as KV pages free up
(vLLM-style): each step() admits pending prompts via ONE batched prefill
(per-token projections fused across sequences; attention per sequence),
then runs ONE fused decode step for every running sequence through the
paged-attention kernel.  Thousands of agent episodes (agents/schedule.py)
gang their LLM turns into these batches; tool round trips run on host
threads between turns, so the GPU stays busy.

Multi-turn agent episodes reuse their KV prefix: a conversation keeps its
sequence alive between turns, the engine rolls the cache back to the shared
prompt prefix and prefills only the delta (observation text) — turn N of an
episode costs O(new tokens), not O(whole transcript).

Sampling: free text decodes greedily (deterministic); grammar decision
tokens sample from the model's masked distribution under a fixed seed
(models/grammar.py) — the whole benchmark stays reproducible.
"""

from __future__ import annotations

import time
from dataclasses import dataclass, field

import torch

from .llama import LlamaModel


@dataclass
class Sequence:
    seq_id: int
    prompt: list[int]            # full prompt tokens for this turn
    max_new_tokens: int
    cached_len: int = 0          # prefix positions already in the KV cache
    out_tokens: list[int] = field(default_factory=list)
    done: bool = False
    keep_alive: bool = False     # conversation: keep KV after completion
    # grammar-constrained decoding (models/grammar.py): the first generated
    # token is logit-masked to `decision_allowed`; if the sampled decision
    # has an entry in `branches`, the rest of the turn is forced to that
    # token script (ending in EOS), else it decodes free.
    decision_allowed: list[int] | None = None
    branches: dict[int, list[int]] | None = None
    script: list[int] | None = None


@dataclass
class EngineStats:
    prefill_tokens: int = 0
    cached_prefix_tokens: int = 0
    decode_tokens: int = 0
    decode_steps: int = 0
    jump_forward_tokens: int = 0
    prefills: int = 0
    prefill_batches: int = 0
    wall_s: float = 0.0
    prefill_s: float = 0.0
    decode_s: float = 0.0


import os as _os
_TIMING = _os.environ.get("QSA_TIMING", "") == "1"
_TUNABLE_DONE = False


def _enable_tunableop() -> None:
    """Load the pre-tuned hipBLASLt/rocBLAS GEMM picks for the decode
    and prefill shapes (PyTorch TunableOp results generated on MI355X by
    tools/tune_gemms.py; +5% on the flagship bench measured A/B).
    Tuning itself stays OFF — unknown shapes fall back to the default
    heuristics; any failure falls back silently."""
    global _TUNABLE_DONE
    if _TUNABLE_DONE or _os.environ.get("QSA_NO_TUNABLEOP") == "1":
        return
    _TUNABLE_DONE = True
    try:
        path = _os.path.join(_os.path.dirname(_os.path.abspath(__file__)),
                             "..", "data", "tunableop_mi355x.csv")
        if _os.path.exists(path):
            t = torch.cuda.tunable
            t.enable(True)
            t.tuning_enable(False)
            t.read_file(path)
    except Exception:
        pass


class Engine:
    def __init__(self, model: LlamaModel, kv_pages: int | None = None,
                 max_batch: int = 256, max_seq_len: int = 4096,
                 eos_id: int | None = None, prefill_batch_tokens: int = 65536,
                 temperature: float = 0.0,
                 valid_vocab: tuple[int, int] | None = None,
                 decision_temperature: float = 1.0):
        if torch.cuda.is_available():
            _enable_tunableop()   # load the pre-tuned GEMM picks
        self.model = model
        # temperature 0 = greedy (the deterministic benchmark contract);
        # > 0 samples via the Gumbel-argmax trick, which stays a single
        # argmax and is hipGraph-capture-safe (torch captures RNG state
        # advancement, so replays draw fresh noise)
        self.temperature = float(temperature)
        # grammar decision tokens are SAMPLED from the model's masked
        # distribution (temperature-1 Gumbel-argmax by default): the
        # model's logits drive tool-vs-finish control flow, as in real
        # tool-choice serving; deterministic under a fixed torch seed.
        # 0 = greedy decisions (degenerate under random-init weights:
        # every episode makes the same choice).
        self.decision_temperature = float(decision_temperature)
        # jump-forward on fully-forced grammar segments (see _admit);
        # QSA_NO_JUMP_FORWARD=1 restores serial decode for comparison
        self.jump_forward = _os.environ.get(
            "QSA_NO_JUMP_FORWARD", "") != "1"
        self.max_batch = max_batch
        self.max_seq_len = max_seq_len
        self.eos_id = eos_id  # None -> run to max_new_tokens (random weights)
        self.prefill_batch_tokens = prefill_batch_tokens
        if kv_pages is None:
            kv_pages = max_batch * ((max_seq_len + 63) // 64) + 64
        self.kv = model.new_kv_cache(kv_pages)
        self._next_id = 0
        self.pending: list[Sequence] = []
        self.running: list[Sequence] = []
        self.finished: list[Sequence] = []   # drained by run_chunk callers
        self.stats = EngineStats()
        # hipGraph-captured decode step (launch-bound otherwise: ~300
        # kernel/GEMM launches per step across 32 layers).  TP ranks
        # capture too — the per-layer RCCL all-reduces are recorded into
        # the graph (validated against eager on hardware,
        # tests/test_gpu_models.py rccl-in-graph test); if capture fails
        # on a given stack the engine falls back to eager decode.
        # TP NOTE: grammar decision sampling draws torch.rand — TP ranks
        # must share a seed so sampled decisions agree across the group.
        self.use_graph = torch.cuda.is_available()
        self._graph = None
        self._graph_failed = False
        self._gbuf: dict = {}
        # free-text sampling mask: tokens outside [lo, hi) (plus EOS) are
        # never emitted — keeps random-init decode inside the tokenizer's
        # trained vocab so outputs are real text (models/grammar.py)
        self._vocab_bias: torch.Tensor | None = None
        if valid_vocab is not None:
            lo, hi = valid_vocab
            bias = torch.full((model.cfg.vocab_size,), float("-inf"),
                              device=model.device, dtype=torch.float32)
            bias[lo:hi] = 0.0
            if self.eos_id is not None:
                bias[self.eos_id] = 0.0
            self._vocab_bias = bias

    def _sample(self, logits: torch.Tensor) -> torch.Tensor:
        """Greedy at temperature 0; Gumbel-argmax sampling otherwise."""
        if self._vocab_bias is not None:
            logits = logits.float() + self._vocab_bias
        if self.temperature <= 0.0:
            return torch.argmax(logits, dim=-1)
        u = torch.rand(logits.shape, device=logits.device,
                       dtype=torch.float32).clamp_min_(1e-20)
        gumbel = -torch.log(-torch.log(u).clamp_min_(1e-20))
        return torch.argmax(logits.float() + self.temperature * gumbel,
                            dim=-1)

    def _sample_first(self, logits: torch.Tensor,
                      admitted: list["Sequence"]) -> list[int]:
        """Sample each admitted sequence's first token, honoring per-row
        decision masks (grammar turns) with ONE host sync."""
        free = self._sample(logits)
        rows = [i for i, s in enumerate(admitted) if s.decision_allowed]
        if not rows:
            return [int(t) for t in free.tolist()]
        K = max(len(admitted[i].decision_allowed) for i in rows)
        idx = torch.tensor(
            [admitted[i].decision_allowed
             + [admitted[i].decision_allowed[-1]]
             * (K - len(admitted[i].decision_allowed)) for i in rows],
            dtype=torch.int64, device=logits.device)
        lens = torch.tensor([len(admitted[i].decision_allowed)
                             for i in rows], device=logits.device)
        pad = torch.arange(K, device=logits.device)[None, :] >= lens[:, None]
        sub = logits[rows].float().gather(1, idx)          # [R, K]
        t = self.decision_temperature
        if t > 0.0:
            u = torch.rand(sub.shape, device=sub.device).clamp_min_(1e-20)
            sub = sub / t + (-torch.log(-torch.log(u)))
        sub = sub.masked_fill(pad, float("-inf"))
        pick = idx.gather(1, torch.argmax(sub, 1, keepdim=True)).squeeze(1)
        out = free.tolist()
        for r, t in zip(rows, pick.tolist()):
            out[r] = int(t)
        return [int(t) for t in out]

    # ---- hipGraph decode -------------------------------------------------
    MAX_RUN = 512  # on-device token-history depth per graph run

    def _ensure_graph(self) -> bool:
        if self._graph is not None:
            return True
        if self._graph_failed:
            return False
        try:
            self._capture_graph()
            return True
        except Exception:
            # capture failed (e.g. an op that refuses stream capture on
            # this stack): permanent eager fallback, correctness first
            self._graph_failed = True
            self.use_graph = False
            self._graph = None
            torch.cuda.synchronize()
            return False

    def _capture_graph(self) -> None:
        dev = self.model.device
        B = self.max_batch
        cap = (self.max_seq_len + 63) // 64
        gb = {
            "tokens": torch.zeros(B, dtype=torch.int64, device=dev),
            "block_table": torch.zeros(B, cap, dtype=torch.int32, device=dev),
            "seq_lens": torch.zeros(B, dtype=torch.int32, device=dev),
            "active": torch.zeros(B, dtype=torch.int32, device=dev),
            "ctr": torch.zeros(1, dtype=torch.int64, device=dev),
            "hist": torch.zeros(self.MAX_RUN, B, dtype=torch.int64, device=dev),
            # grammar scripts: forced token at run-step j for each row
            # (script_mask False -> free sample)
            "script": torch.zeros(B, self.MAX_RUN, dtype=torch.int64,
                                  device=dev),
            "script_mask": torch.zeros(B, self.MAX_RUN, dtype=torch.bool,
                                       device=dev),
        }

        def body():
            # self-feeding decode step: advance seq_lens in-graph, append the
            # step's k/v at seq_lens-1, attend, sample, record, feed back.
            gb["seq_lens"].add_(gb["active"])
            positions = (gb["seq_lens"] - 1).clamp(min=0).int()
            logits = self.model.forward_decode(
                gb["tokens"], self.kv, gb["block_table"], gb["seq_lens"],
                positions)
            nxt = self._sample(logits)
            # grammar-forced tokens override the free sample in-graph
            j = gb["ctr"].clamp(max=self.MAX_RUN - 1)
            forced = gb["script"].index_select(1, j).squeeze(1)
            fmask = gb["script_mask"].index_select(1, j).squeeze(1)
            nxt = torch.where(fmask, forced, nxt)
            gb["hist"].index_copy_(0, gb["ctr"], nxt.unsqueeze(0))
            gb["ctr"].add_(1)
            gb["tokens"].copy_(nxt)

        # warm up the exact captured ops (allocator + rocBLAS algo selection)
        for _ in range(2):
            body()
        gb["ctr"].zero_()
        gb["seq_lens"].zero_()
        torch.cuda.synchronize()
        g = torch.cuda.CUDAGraph()
        with torch.cuda.graph(g):
            body()
        self._graph = g
        self._gbuf = gb

    def _decode_run_eager(self, batch: list[Sequence], run: int) -> None:
        """Eager fallback for a graph run (capture unavailable)."""
        for _ in range(run):
            live = [s for s in batch if not s.done]
            if not live:
                break
            for s in live:
                self.kv.extend(s.seq_id, len(s.prompt) + len(s.out_tokens))
            self._decode_step_eager(live)

    def _decode_step_eager(self, batch: list[Sequence]) -> None:
        dev = self.model.device
        tokens = torch.tensor([s.out_tokens[-1] for s in batch],
                              dtype=torch.int64, device=dev)
        positions = torch.tensor(
            [len(s.prompt) + len(s.out_tokens) - 1 for s in batch],
            dtype=torch.int32, device=dev)
        seq_ids = [s.seq_id for s in batch]
        bt = self.kv.block_table(seq_ids)
        sl = self.kv.seq_lens_tensor(seq_ids)
        logits = self.model.forward_decode(tokens, self.kv, bt, sl,
                                           positions)
        nxt = self._sample(logits).tolist()
        for s, tok in zip(batch, nxt):
            if s.script:
                # grammar-forced continuation (script index: out_tokens[0]
                # was the decision token)
                si = len(s.out_tokens) - 1
                if si < len(s.script):
                    tok = s.script[si]
            s.out_tokens.append(int(tok))
            self._maybe_finish(s)
        self.stats.decode_tokens += len(batch)
        self.stats.decode_steps += 1

    def _decode_run_graph(self, batch: list[Sequence], run: int) -> None:
        """Run `run` decode steps for `batch` with zero host round trips:
        pages are pre-extended, the graph self-feeds, one sync at the end."""
        if not self._ensure_graph():
            self._decode_run_eager(batch, run)
            return
        self._graph_run_begin(batch, run)
        self._graph_run_end(batch, run)

    def _graph_run_begin(self, batch: list[Sequence], run: int) -> None:
        """Fill the graph buffers and queue `run` replays (async on the
        current stream — no host sync; _graph_run_end collects)."""
        gb = self._gbuf
        n = len(batch)
        dev = self.model.device
        for s in batch:  # pre-extend pages for the whole run
            self.kv.extend(s.seq_id, len(s.prompt) + len(s.out_tokens) + run)
            # seq_lens the graph starts from EXCLUDE the token the first
            # replay decodes: the in-graph add_(active) then yields
            # len(prompt)+len(out) for step 1 — the same "includes the
            # new token" value the eager path passes (appending at slot
            # len(prompt)+len(out)-1, no zero hole after the prompt)
            self.kv._seq_len[s.seq_id] = \
                len(s.prompt) + len(s.out_tokens) - 1
        seq_ids = [s.seq_id for s in batch]
        bt = self.kv.block_table(seq_ids)
        gb["block_table"][:n, :bt.shape[1]] = bt
        gb["seq_lens"][:n] = self.kv.seq_lens_tensor(seq_ids)
        gb["tokens"][:n] = torch.tensor([s.out_tokens[-1] for s in batch],
                                        dtype=torch.int64, device=dev)
        gb["active"][:n] = 1
        if n < self.max_batch:
            gb["seq_lens"][n:] = 0
            gb["active"][n:] = 0
        # grammar scripts for this run window: run-step j emits
        # out_tokens[d + j] whose script index is d - 1 + j (out_tokens[0]
        # was the prefill-sampled decision token)
        if any(s.script for s in batch):
            sc = torch.zeros(n, run, dtype=torch.int64)
            sm = torch.zeros(n, run, dtype=torch.bool)
            for i, s in enumerate(batch):
                if not s.script:
                    continue
                d = len(s.out_tokens)
                window = s.script[d - 1: d - 1 + run]
                if window:
                    sc[i, :len(window)] = torch.tensor(window,
                                                       dtype=torch.int64)
                    sm[i, :len(window)] = True
            gb["script"][:n, :run] = sc.to(dev)
            gb["script_mask"][:n, :run] = sm.to(dev)
            gb["script_mask"][:n, run:] = False
            if n < self.max_batch:
                gb["script_mask"][n:] = False
        else:
            gb["script_mask"].fill_(False)
        gb["ctr"].zero_()
        if _TIMING:
            torch.cuda.synchronize()
            self._t_graph0 = time.perf_counter()
        for _ in range(run):
            self._graph.replay()

    def _graph_run_end(self, batch: list[Sequence], run: int) -> None:
        gb = self._gbuf
        n = len(batch)
        if _TIMING:
            torch.cuda.synchronize()
            self.stats.decode_s += time.perf_counter() - self._t_graph0
        hist = gb["hist"][:run, :n].t().tolist()  # one sync
        for s, toks in zip(batch, hist):
            new = [int(t) for t in toks]
            if self.eos_id is not None and self.eos_id in new:
                # honor mid-run EOS: keep tokens up to and including it
                new = new[: new.index(self.eos_id) + 1]
            s.out_tokens.extend(new)
            self.kv._seq_len[s.seq_id] = len(s.prompt) + len(s.out_tokens)
            self._maybe_finish(s)
        self.stats.decode_tokens += n * run
        self.stats.decode_steps += run

    # ------------------------------------------------------------------
    def submit(self, prompt_tokens: list[int], max_new_tokens: int,
               continue_from: Sequence | None = None,
               keep_alive: bool = False, constraint=None) -> Sequence:
        """Queue a prompt.  continue_from: a completed keep-alive sequence
        whose prompt is a prefix of this one — its KV prefix is reused.
        constraint: models.grammar.CompiledGrammar for this turn."""
        prompt = list(prompt_tokens[: self.max_seq_len - 1])
        decision_allowed = branches = None
        if constraint is not None:
            decision_allowed = list(constraint.decision_allowed)
            branches = dict(constraint.branches)
            # a forced tool-call script must fit in the turn
            max_new_tokens = max(max_new_tokens,
                                 constraint.max_script_len() + 1)
            max_new_tokens = min(max_new_tokens, self.MAX_RUN)
        # prompt + decode must fit the sequence budget (KV pages AND the
        # model's rope table)
        max_new_tokens = max(1, min(max_new_tokens,
                                    self.max_seq_len - len(prompt)))
        if continue_from is not None and not continue_from.keep_alive:
            continue_from = None
        if continue_from is not None:
            # at least one token must prefill (last-position logits)
            shared = min(len(continue_from.prompt), len(prompt) - 1)
            # shared prefix = the previous turn's prompt (its decode-token KV
            # is rolled back); bail to fresh prefill on any mismatch
            if prompt[:shared] != continue_from.prompt[:shared]:
                shared = 0
            if shared > 0:
                seq = Sequence(continue_from.seq_id, prompt, max_new_tokens,
                               cached_len=shared, keep_alive=keep_alive,
                               decision_allowed=decision_allowed,
                               branches=branches)
                self.kv.truncate(seq.seq_id, shared)
                self._next_id = max(self._next_id, seq.seq_id + 1)
                self.pending.append(seq)
                return seq
            self.kv.free(continue_from.seq_id)
        seq = Sequence(self._next_id, prompt, max_new_tokens,
                       keep_alive=keep_alive,
                       decision_allowed=decision_allowed, branches=branches)
        self._next_id += 1
        self.pending.append(seq)
        return seq

    def release(self, seq: Sequence) -> None:
        """Free a keep-alive conversation's KV."""
        if seq.keep_alive:
            seq.keep_alive = False
            self.kv.free(seq.seq_id)

    # ------------------------------------------------------------------
    def _admit(self) -> None:
        # group like-sized prefills: padded-batch attention pads every
        # admitted sequence to the batch max delta, so sort pending by
        # (has-cached-prefix, delta desc) — fresh long prompts batch with
        # each other, short continuation deltas (large ctx, tiny delta)
        # batch together, minimizing both pad waste and KV-gather width
        self.pending.sort(
            key=lambda s: (s.cached_len > 0,
                           -(len(s.prompt) - s.cached_len)))
        batch_items: list[tuple[torch.Tensor, int, int]] = []
        admitted: list[Sequence] = []
        new_tokens = 0
        while self.pending and len(self.running) + len(admitted) < self.max_batch:
            seq = self.pending[0]
            total = len(seq.prompt) + seq.max_new_tokens
            have = len(self.kv._seq_pages.get(seq.seq_id, ())) \
                if seq.cached_len else 0
            need = self.kv.pages_for(total) - have
            if need > self.kv.free_pages:
                break
            delta = len(seq.prompt) - seq.cached_len
            if batch_items and new_tokens + delta > self.prefill_batch_tokens:
                break
            self.pending.pop(0)
            if seq.cached_len:
                self.kv.extend(seq.seq_id, len(seq.prompt))
            else:
                self.kv.allocate(seq.seq_id, len(seq.prompt))
            toks = torch.tensor(seq.prompt[seq.cached_len:],
                                dtype=torch.int64, device=self.model.device)
            batch_items.append((toks, seq.seq_id, seq.cached_len))
            admitted.append(seq)
            new_tokens += delta
            self.stats.prefill_tokens += delta
            self.stats.cached_prefix_tokens += seq.cached_len
            self.stats.prefills += 1
        if not admitted:
            return
        if _TIMING:
            torch.cuda.synchronize()
            _t0 = time.perf_counter()
        logits = self.model.forward_prefill_batch(batch_items, self.kv)
        if _TIMING:
            torch.cuda.synchronize()
            self.stats.prefill_s += time.perf_counter() - _t0
        first = self._sample_first(logits, admitted)
        self.stats.prefill_batches += 1
        for seq, tok in zip(admitted, first):
            seq.out_tokens.append(int(tok))
            if seq.branches is not None:
                seq.script = seq.branches.get(int(tok))
                if seq.script is not None and self.jump_forward:
                    # jump-forward decoding (the SGLang fast-forward on
                    # grammar-forced segments): once the sampled decision
                    # commits to a tool branch, EVERY remaining token of
                    # the turn is grammar-forced — no model choice is
                    # left, and the forced tokens' KV is never reused
                    # (the next turn's prompt carries the OBSERVATION,
                    # not the raw call text, and retire rolls the cache
                    # back to the prompt) — so the engine emits the
                    # script directly instead of serial decode steps.
                    room = seq.max_new_tokens - 1
                    seq.out_tokens.extend(seq.script[:room])
                    self.stats.jump_forward_tokens += min(
                        len(seq.script), room)
            self.running.append(seq)
            self._maybe_finish(seq)

    def _maybe_finish(self, seq: Sequence) -> None:
        if len(seq.out_tokens) >= seq.max_new_tokens or (
                self.eos_id is not None and seq.out_tokens
                and seq.out_tokens[-1] == self.eos_id):
            seq.done = True

    def _retire(self) -> None:
        done = [s for s in self.running if s.done]
        for s in done:
            if not s.keep_alive:
                self.kv.free(s.seq_id)
            else:
                # roll back to the prompt: the next turn extends the prompt,
                # not the decoded tokens
                self.kv.truncate(s.seq_id, len(s.prompt))
        self.finished.extend(done)
        self.running = [s for s in self.running if not s.done]

    def run_chunk(self, max_run: int = 16) -> int:
        """One continuous-batching slice: admit whatever fits, then up
        to `max_run` decode steps (one graph run).  Returns remaining
        work count.  The event-driven scheduler
        (agents/schedule.py run_episodes_continuous) interleaves these
        slices with tool I/O completions, so late turns join the running
        batch instead of waiting for a global round barrier."""
        # NOTE: an overlapped variant (admissions on a side HIP stream
        # while the decode graph replays) was measured to hang
        # NON-deterministically at bench scale — the prefill's H2D /
        # allocator traffic races the replaying graph's memory pool on
        # this stack — so admissions and decode run sequentially.
        while self.pending:
            before = len(self.pending)
            self._admit()
            if len(self.pending) == before:
                break
        batch = [s for s in self.running if not s.done]
        if batch:
            run = min(min(s.max_new_tokens - len(s.out_tokens)
                          for s in batch), max_run, self.MAX_RUN)
            if run > 0:
                if self.use_graph:
                    self._decode_run_graph(batch, run)
                else:
                    self._decode_run_eager(batch, run)
        self._retire()
        if self.pending and not self.running and not batch:
            raise MemoryError("decode stalled: pending prompts cannot be "
                              "admitted (KV pages exhausted?)")
        return len(self.running) + len(self.pending)

    def step(self) -> int:
        """Admit + one decode step; returns remaining work count."""
        self._admit()
        batch = [s for s in self.running if not s.done]
        if not batch:
            self._retire()
            return len(self.running) + len(self.pending)
        for s in batch:
            self.kv.extend(s.seq_id, len(s.prompt) + len(s.out_tokens))
        if self.use_graph:
            # one-step graph run (appends + finishes internally; falls
            # back to eager if capture is unavailable)
            self._decode_run_graph(batch, 1)
        else:
            self._decode_step_eager(batch)
        self._retire()
        return len(self.running) + len(self.pending)

    def run_to_completion(self) -> None:
        self.finished.clear()   # batch callers don't drain it
        t0 = time.perf_counter()
        if self.use_graph:
            while self.pending or self.running:
                before = len(self.pending)
                # drain the whole pending queue (possibly several prefill
                # batches) before decoding: decode runs want the full batch
                while self.pending:
                    n_before = len(self.pending)
                    self._admit()
                    if len(self.pending) == n_before:
                        break
                batch = [s for s in self.running if not s.done]
                if not batch:
                    self._retire()
                    if self.pending and len(self.pending) == before and \
                            not self.running:
                        raise MemoryError(
                            "decode stalled: pending prompts cannot be "
                            "admitted (KV pages exhausted?)")
                    continue
                run = min(min(s.max_new_tokens - len(s.out_tokens)
                              for s in batch), self.MAX_RUN)
                if run > 0:
                    self._decode_run_graph(batch, run)
                self._retire()
        else:
            while self.pending or self.running:
                n_pending = len(self.pending)
                remaining = self.step()
                if remaining and not self.running and \
                        len(self.pending) == n_pending:
                    raise MemoryError(
                        "decode stalled: pending prompts cannot be "
                        "admitted (KV pages exhausted?)")
        if torch.cuda.is_available():
            torch.cuda.synchronize()
        self.stats.wall_s += time.perf_counter() - t0

    def generate_batch(self, prompts: list[list[int]],
                       max_new_tokens: list[int]) -> list[list[int]]:
        seqs = [self.submit(p, m) for p, m in zip(prompts, max_new_tokens)]
        self.run_to_completion()
        return [s.out_tokens for s in seqs]


class EngineLLM:
    """llm_batch adapter for the lab pipelines: text in, text out.

    When the scheduler passes conversation ids (agents/schedule.py), each
    conversation keeps its engine sequence alive between turns and only the
    prompt delta is prefassed (KV prefix reuse).  Per-turn TurnGrammar
    specs (models/grammar.py) are compiled against the tokenizer and
    passed into the engine as decision masks + forced scripts."""

    def __init__(self, engine: Engine, tokenizer=None):
        if tokenizer is None:
            from .tokenizer import default_tokenizer
            tokenizer = default_tokenizer(engine.model.cfg.vocab_size)
        self.engine = engine
        self.tokenizer = tokenizer
        self._convs: dict = {}
        self._by_seq: dict = {}

    def __call__(self, prompts: list[str], max_new_tokens: list[int],
                 conv_ids: list | None = None,
                 grammars: list | None = None) -> list[str]:
        from .grammar import compile_grammar
        seqs = []
        for i, (p, m) in enumerate(zip(prompts, max_new_tokens)):
            conv = conv_ids[i] if conv_ids is not None else None
            prev = self._convs.get(conv) if conv is not None else None
            g = grammars[i] if grammars is not None else None
            constraint = compile_grammar(g, self.tokenizer) \
                if g is not None else None
            enc = self.tokenizer.encode(p)
            seq = self.engine.submit(enc, m, continue_from=prev,
                                     keep_alive=conv is not None,
                                     constraint=constraint)
            if conv is not None:
                self._convs[conv] = seq
            seqs.append(seq)
        self.engine.run_to_completion()
        return [self.tokenizer.decode(s.out_tokens) for s in seqs]

    # ---- event-driven (continuous) scheduling interface ----------------
    def submit_turn(self, conv_id, prompt: str, max_new_tokens: int,
                    grammar=None) -> None:
        """Queue one agent turn WITHOUT running the engine — the
        continuous scheduler (agents/schedule.py) interleaves engine
        slices with tool I/O so turns join the running batch as they
        become ready."""
        from .grammar import compile_grammar
        constraint = compile_grammar(grammar, self.tokenizer)             if grammar is not None else None
        prev = self._convs.get(conv_id)
        enc = self.tokenizer.encode(prompt)
        seq = self.engine.submit(enc, max_new_tokens, continue_from=prev,
                                 keep_alive=True, constraint=constraint)
        self._convs[conv_id] = seq
        self._by_seq[id(seq)] = conv_id

    def pop_finished(self) -> list[tuple]:
        """Drain (conv_id, decoded text) for turns completed since the
        last call (engine.finished is filled by retire)."""
        out = []
        for seq in self.engine.finished:
            cid = self._by_seq.pop(id(seq), None)
            if cid is not None:
                out.append((cid, self.tokenizer.decode(seq.out_tokens)))
        self.engine.finished.clear()
        return out

    def release(self, conv_id) -> None:
        seq = self._convs.pop(conv_id, None)
        if seq is not None:
            self._by_seq.pop(id(seq), None)
            self.engine.release(seq)

    def release_all(self) -> None:
        for conv in list(self._convs):
            self.release(conv)
