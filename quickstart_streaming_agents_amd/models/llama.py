"""Llama-family decoder for the agent LLM (K4), MI355X-native.

The reference's textgen model is a managed Bedrock/Azure endpoint
(terraform/core/main.tf:461,495); here it is an on-GPU decoder built from
the hand-written HIP kernels (rmsnorm / rope / swiglu / paged-attention
decode) plus rocBLAS GEMMs via torch.nn.functional.linear for the plain
projections.  Weights are random-init bf16 (the north-star benchmark runs
synthetic data / random weights; the architecture is the named config).

Decode path (the hot path): fully fused per layer —
  rmsnorm_residual -> merged qkv GEMM -> rope_inplace -> kv_append ->
  paged_attn_decode -> o GEMM -> rmsnorm_residual -> merged gate_up GEMM ->
  swiglu -> down GEMM
Prefill: per-sequence chunk with GEMM scores + the causal softmax kernel.
"""

from __future__ import annotations

from dataclasses import dataclass

import torch
import torch.nn.functional as F

from ..ops import ext
from .kv_cache import PagedKVCache


@dataclass
class LlamaConfig:
    name: str = "llama3-8b"
    vocab_size: int = 128_256
    hidden: int = 4096
    n_layers: int = 32
    n_q_heads: int = 32
    n_kv_heads: int = 8
    d_head: int = 128
    ffn: int = 14_336
    rope_theta: float = 500_000.0
    max_pos: int = 8192
    norm_eps: float = 1e-5

    @classmethod
    def preset(cls, name: str) -> "LlamaConfig":
        if name in ("llama3-8b", "llama-3-8b"):
            return cls()
        if name == "llama3-1b":
            return cls(name=name, vocab_size=128_256, hidden=2048, n_layers=16,
                       n_q_heads=32, n_kv_heads=8, d_head=64, ffn=8192)
        if name == "tiny":
            return cls(name=name, vocab_size=2048, hidden=256, n_layers=2,
                       n_q_heads=4, n_kv_heads=1, d_head=64, ffn=512,
                       max_pos=2048)
        raise ValueError(f"unknown preset {name}")


class LlamaModel:
    def __init__(self, cfg: LlamaConfig, device: str = "cuda",
                 dtype=torch.bfloat16, seed: int = 0):
        self.cfg = cfg
        self.device = device
        self.dtype = dtype
        gen_dev = device if str(device).startswith("cuda") else "cpu"
        g = torch.Generator(device=gen_dev).manual_seed(seed)

        def w(*shape, std=0.02):
            # generate on-device: 8B-class random init in seconds, no 32 GB
            # host spike
            t = torch.randn(*shape, generator=g, dtype=torch.float32,
                            device=gen_dev) * std
            return t.to(device=device, dtype=dtype)

        c = cfg
        self.qkv_dim = (c.n_q_heads + 2 * c.n_kv_heads) * c.d_head
        self.embed = w(c.vocab_size, c.hidden)
        self.lm_head = w(c.vocab_size, c.hidden)
        self.final_norm = torch.ones(c.hidden, device=device, dtype=dtype)
        self.layers = []
        for _ in range(c.n_layers):
            self.layers.append({
                "attn_norm": torch.ones(c.hidden, device=device, dtype=dtype),
                "mlp_norm": torch.ones(c.hidden, device=device, dtype=dtype),
                "wqkv": w(self.qkv_dim, c.hidden),
                "wo": w(c.hidden, c.n_q_heads * c.d_head),
                "wgu": w(2 * c.ffn, c.hidden),
                "wdown": w(c.hidden, c.ffn, std=0.02 / (2 * c.n_layers) ** 0.5),
            })
        # rope tables
        half = c.d_head // 2
        inv = 1.0 / (c.rope_theta **
                     (torch.arange(half, dtype=torch.float64) / half))
        ang = torch.outer(torch.arange(c.max_pos, dtype=torch.float64), inv)
        self.rope_cos = ang.cos().float().to(device)
        self.rope_sin = ang.sin().float().to(device)
        self.scale = 1.0 / (c.d_head ** 0.5)

    def new_kv_cache(self, n_pages: int) -> PagedKVCache:
        return PagedKVCache(self.cfg.n_layers, self.cfg.n_kv_heads,
                            self.cfg.d_head, n_pages, self.device, self.dtype)

    # ------------------------------------------------------------------
    def _split_qkv(self, qkv: torch.Tensor, n: int):
        c = self.cfg
        qd = c.n_q_heads * c.d_head
        kd = c.n_kv_heads * c.d_head
        q = qkv[:, :qd].reshape(n, c.n_q_heads, c.d_head).contiguous()
        k = qkv[:, qd:qd + kd].reshape(n, c.n_kv_heads, c.d_head).contiguous()
        v = qkv[:, qd + kd:].reshape(n, c.n_kv_heads, c.d_head).contiguous()
        return q, k, v

    @torch.no_grad()
    def forward_decode(self, tokens: torch.Tensor, kv: PagedKVCache,
                       block_table: torch.Tensor, seq_lens: torch.Tensor,
                       positions: torch.Tensor) -> torch.Tensor:
        """tokens [B] i64 -> logits [B, vocab].  seq_lens INCLUDE the new
        token (its k/v is appended at position seq_lens-1)."""
        c = self.cfg
        B = tokens.shape[0]
        e = ext()
        res = self.embed.index_select(0, tokens).contiguous()
        h = None
        mlp_out = None
        for li, L in enumerate(self.layers):
            if li == 0:
                h = e.rmsnorm(res, L["attn_norm"], c.norm_eps)
            else:
                h = e.rmsnorm_residual(mlp_out, res, L["attn_norm"], c.norm_eps)
            qkv = F.linear(h, L["wqkv"])
            # strided [B, H, D] views straight into the fused qkv buffer —
            # the kernels take row strides, no contiguous() copies
            qd = c.n_q_heads * c.d_head
            kd = c.n_kv_heads * c.d_head
            q = qkv[:, :qd].view(B, c.n_q_heads, c.d_head)
            k = qkv[:, qd:qd + kd].view(B, c.n_kv_heads, c.d_head)
            v = qkv[:, qd + kd:].view(B, c.n_kv_heads, c.d_head)
            e.rope_inplace(q, k, self.rope_cos, self.rope_sin, positions)
            e.kv_append(k, v, kv.k[li], kv.v[li], block_table, seq_lens)
            attn = e.paged_attn_decode(q, kv.k[li], kv.v[li], block_table,
                                       seq_lens, self.scale)
            o = F.linear(attn.view(B, -1), L["wo"])
            h = e.rmsnorm_residual(o, res, L["mlp_norm"], c.norm_eps)
            gu = F.linear(h, L["wgu"])
            act = e.swiglu(gu[:, :c.ffn], gu[:, c.ffn:])
            mlp_out = F.linear(act, L["wdown"])
        final_h = e.rmsnorm_residual(mlp_out, res, self.final_norm, c.norm_eps)
        return F.linear(final_h, self.lm_head)

    def _gather_kv(self, kc_l: torch.Tensor, vc_l: torch.Tensor,
                   pages: list[int], n: int):
        """Gather positions [0, n) of a sequence from the paged cache ->
        (K [KVH, n, D], V [KVH, n, D]) contiguous for the prefill GEMMs."""
        c = self.cfg
        idx = torch.tensor(pages, dtype=torch.int64, device=self.device)
        kpg = kc_l.index_select(0, idx)        # [np, KVH, D/8, 64, 8]
        kseq = kpg.permute(1, 0, 3, 2, 4).reshape(
            c.n_kv_heads, len(pages) * 64, c.d_head)[:, :n]
        vpg = vc_l.index_select(0, idx)        # [np, KVH, 64, D]
        vseq = vpg.permute(1, 0, 2, 3).reshape(
            c.n_kv_heads, len(pages) * 64, c.d_head)[:, :n]
        return kseq, vseq

    @torch.no_grad()
    def forward_prefill_batch(self, items: list[tuple[torch.Tensor, int, int]],
                              kv: PagedKVCache) -> torch.Tensor:
        """Batched (and incremental) prefill.

        items: (new_tokens [t_i] i64, seq_id, start_pos).  start_pos > 0 means
        the sequence's first start_pos positions are already in the paged
        cache (prefix reuse across agent turns).  All per-token projections
        run as ONE fused GEMM over the concatenation; only attention loops
        per sequence.  Returns last-position logits [n_items, vocab].
        """
        c = self.cfg
        e = ext()
        dev = self.device
        lens = [int(t.shape[0]) for t, _, _ in items]
        offs = [0]
        for t in lens:
            offs.append(offs[-1] + t)
        T = offs[-1]
        tokens = torch.cat([t for t, _, _ in items])
        positions = torch.cat([
            torch.arange(s, s + n, dtype=torch.int32, device=dev)
            for (_, _, s), n in zip(items, lens)]).contiguous()
        slots = torch.cat([
            kv.slot_ids(sid, s, n)
            for (_, sid, s), n in zip(items, lens)]).contiguous()

        res = self.embed.index_select(0, tokens).contiguous()
        h = None
        mlp_out = None
        for li, L in enumerate(self.layers):
            if li == 0:
                h = e.rmsnorm(res, L["attn_norm"], c.norm_eps)
            else:
                h = e.rmsnorm_residual(mlp_out, res, L["attn_norm"], c.norm_eps)
            qkv = F.linear(h, L["wqkv"])
            q, k, v = self._split_qkv(qkv, T)
            e.rope_inplace(q, k, self.rope_cos, self.rope_sin, positions)
            e.kv_scatter(k, v, kv.k[li], kv.v[li], slots)
            attn = torch.empty(T, c.n_q_heads * c.d_head, dtype=self.dtype,
                               device=dev)
            for (tt, sid, start), n, off in zip(items, lens, offs):
                ctx = start + n
                if start == 0:
                    kseq = k[off:off + n].permute(1, 0, 2)
                    vseq = v[off:off + n].permute(1, 0, 2)
                else:
                    kseq, vseq = self._gather_kv(kv.k[li], kv.v[li],
                                                 kv._seq_pages[sid], ctx)
                # GQA grouped: [KVH, R*n, D] x [KVH, ctx, D]; bf16 MFMA
                # GEMMs (f32 bmm is 1/16 the MFMA rate on CDNA4)
                R = c.n_q_heads // c.n_kv_heads
                qf = q[off:off + n].permute(1, 0, 2).reshape(
                    c.n_kv_heads, R * n, c.d_head)
                kf = kseq.contiguous()                         # [KVH, ctx, D]
                vf = vseq.contiguous()
                scores = (torch.bmm(qf, kf.transpose(1, 2)).float()
                          * self.scale) \
                    .reshape(c.n_q_heads * n, ctx).contiguous()
                e.softmax_rows_(scores, start, True, n, None)
                probs = scores.reshape(c.n_kv_heads, R * n, ctx) \
                    .to(self.dtype)
                a = torch.bmm(probs, vf)
                attn[off:off + n] = a.reshape(c.n_q_heads, n, c.d_head) \
                    .permute(1, 0, 2).reshape(n, -1)
            o = F.linear(attn, L["wo"])
            h = e.rmsnorm_residual(o, res, L["mlp_norm"], c.norm_eps)
            gu = F.linear(h, L["wgu"])
            act = e.swiglu(gu[:, :c.ffn], gu[:, c.ffn:])
            mlp_out = F.linear(act, L["wdown"])
        final_h = e.rmsnorm_residual(mlp_out, res, self.final_norm, c.norm_eps)
        last = torch.tensor([offs[i] + lens[i] - 1 for i in range(len(items))],
                            dtype=torch.int64, device=dev)
        return F.linear(final_h.index_select(0, last), self.lm_head)

    @torch.no_grad()
    def forward_prefill(self, tokens: torch.Tensor, kv: PagedKVCache,
                        seq_id: int) -> torch.Tensor:
        """Single-sequence prefill (wrapper over the batch path)."""
        return self.forward_prefill_batch([(tokens, seq_id, 0)], kv)[0]
