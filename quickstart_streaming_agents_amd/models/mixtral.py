"""Mixtral-8x7B MoE decoder (BASELINE.json config 5), MI355X-native.

Same attention stack as the Llama decoder (paged-attention HIP kernel,
GQA 32/8 heads, rope) — the FFN is a routed mixture: top-2 of 8 SwiGLU
experts, softmax-renormalized gates (the Mixtral recipe).  Experts shard
across the EP group via RCCL all-to-all over xGMI (parallel/ep.py); with
288 GB HBM3E per GPU the whole 8x7B fits on ONE GPU (bf16 ~93 GB), so
EP=1 single-GPU serving works too and EP>1 trades capacity for FFN
throughput on bursty streaming batches.

Attention TP and MoE EP use the same process group (config 5: "TP+EP
across 8 MI355X"): q/kv heads shard tp_size-ways while each rank owns
n_experts/ep_size full experts.
"""

from __future__ import annotations

from dataclasses import dataclass

import torch
import torch.nn.functional as F

from ..ops import dispatch as D
from ..parallel.ep import ExpertDispatch, all_gather_rows
from .llama import LlamaConfig, LlamaModel


@dataclass
class MixtralConfig(LlamaConfig):
    name: str = "mixtral-8x7b"
    vocab_size: int = 32_000
    hidden: int = 4096
    n_layers: int = 32
    n_q_heads: int = 32
    n_kv_heads: int = 8
    d_head: int = 128
    ffn: int = 14_336
    rope_theta: float = 1_000_000.0
    n_experts: int = 8
    top_k: int = 2

    @classmethod
    def preset(cls, name: str) -> "MixtralConfig":
        if name in ("mixtral-8x7b", "mixtral"):
            return cls()
        if name == "tiny-moe":
            return cls(name=name, vocab_size=2048, hidden=256, n_layers=2,
                       n_q_heads=4, n_kv_heads=2, d_head=64, ffn=512,
                       max_pos=2048, n_experts=4, top_k=2)
        if name == "mixtral-8x22b":
            # ~141B params: ~282 GB bf16 — does NOT fit one 288 GB GPU
            # with KV cache; serve with EP>=2 (parallel/ep.py shards the
            # 8 experts across ranks, each rank ~46 GB of experts +
            # ~16 GB shared)
            return cls(name=name, vocab_size=32_768, hidden=6144,
                       n_layers=56, n_q_heads=48, n_kv_heads=8,
                       d_head=128, ffn=16_384, rope_theta=1_000_000.0,
                       max_pos=8192, n_experts=8, top_k=2)
        raise ValueError(f"unknown preset {name}")


class MixtralModel(LlamaModel):
    def __init__(self, cfg: MixtralConfig, device: str = "cuda",
                 dtype=torch.bfloat16, seed: int = 0,
                 tp_rank: int = 0, tp_size: int = 1, tp_group=None,
                 ep_size: int | None = None):
        self._moe_cfg = cfg
        ep_size = tp_size if ep_size is None else ep_size
        assert cfg.n_experts % ep_size == 0, "experts must divide EP"
        self.ep_size = ep_size
        self._dispatch = None  # built after super().__init__ sets tp_rank
        super().__init__(cfg, device=device, dtype=dtype, seed=seed,
                         tp_rank=tp_rank, tp_size=tp_size, tp_group=tp_group)
        self._dispatch = ExpertDispatch(cfg.n_experts, ep_size,
                                        tp_rank % ep_size, tp_group)

    def _ffn_weights(self, w, rows, cols) -> dict:
        """Router (replicated) + this rank's experts (FULL weights under
        EP — no TP slice inside an expert).  Every rank generates every
        expert's tensors to keep the generator stream identical, then
        keeps only its own (the others are freed immediately)."""
        c = self._moe_cfg
        router = w(c.n_experts, c.hidden)
        per_rank = c.n_experts // self.ep_size
        my_lo = (self.tp_rank % self.ep_size) * per_rank
        experts = {}
        packed = {}
        pack = (str(self.device).startswith("cuda")
                and self.dtype == torch.bfloat16)
        for e in range(c.n_experts):
            wgu = w(2 * c.ffn, c.hidden)
            wdown = w(c.hidden, c.ffn, std=0.02 / (2 * c.n_layers) ** 0.5)
            if my_lo <= e < my_lo + per_rank or self.ep_size == 1:
                experts[e] = (wgu, wdown)
                if pack and D.can_pack_weight(*wgu.shape) and \
                        D.can_pack_weight(*wdown.shape) and \
                        wgu.numel() <= 6144 * 4096:
                    packed[e] = (D.pack_weight_frag(wgu),
                                 D.pack_weight_frag(wdown))
            else:
                del wgu, wdown
        return {"router": router, "experts": experts,
                "experts_f": packed}

    def _ffn(self, L: dict, h: torch.Tensor) -> torch.Tensor:
        c = self._moe_cfg
        logits = F.linear(h, L["router"]).float()
        probs = torch.softmax(logits, dim=-1)
        top_w, top_idx = torch.topk(probs, c.top_k, dim=-1)
        top_w = top_w / top_w.sum(dim=-1, keepdim=True)

        def expert_fn(e: int, rows: torch.Tensor) -> torch.Tensor:
            wgu, wdown = L["experts"][e]
            pk = L.get("experts_f", {}).get(e)
            if pk is not None and rows.is_cuda and rows.shape[0] <= 32:
                gu = D.skinny_linear(rows, pk[0], *wgu.shape)
                act = D.swiglu(gu[:, :c.ffn], gu[:, c.ffn:])
                return D.skinny_linear(act, pk[1], *wdown.shape)
            gu = F.linear(rows, wgu)
            act = D.swiglu(gu[:, :c.ffn], gu[:, c.ffn:])
            return F.linear(act, wdown)

        if self.ep_size == 1:
            if h.shape[0] <= 64 and h.is_cuda:
                # decode-sized batch: dense compute-all-experts.  Top-2/8 on
                # 24+ tokens touches every expert's weights anyway (the MoE
                # FFN is weight-bandwidth-bound, FLOPs are incidental), and
                # this path is hipGraph-capture-safe: no .nonzero()/host
                # sync, routing becomes a masked weighted sum.
                gates = torch.zeros_like(probs)  # [T, E]
                gates.scatter_(1, top_idx, top_w)
                out = torch.zeros(h.shape, dtype=torch.float32,
                                  device=h.device)
                for e in range(c.n_experts):
                    y = expert_fn(e, h)
                    out += gates[:, e:e + 1] * y.float()
                return out.to(h.dtype)
            return self._dispatch.run(h, top_idx, top_w, expert_fn)
        # TP group == EP group: h is identical on all ranks after the
        # attention all-reduce, so each rank dispatches only its token
        # slice (no redundant expert compute) and the slices all-gather.
        T = h.shape[0]
        r = self.tp_rank % self.ep_size
        lo, hi = r * T // self.ep_size, (r + 1) * T // self.ep_size
        out_local = self._dispatch.run(h[lo:hi], top_idx[lo:hi],
                                       top_w[lo:hi], expert_fn)
        return all_gather_rows(out_local, self.tp_group)
