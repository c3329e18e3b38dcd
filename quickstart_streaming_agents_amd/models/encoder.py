"""On-GPU embedding encoder (K1): bge-small-class transformer -> 1536-d.

Replaces the managed embedding models (titan-embed-v1 / ada-002, both
1536-d — terraform/core/main.tf:529,563; dims contract validate.py:56-62).
A bge-small-class bidirectional encoder (12 layers, 384 hidden, attention
block 8 heads x 64 = 512 so the head geometry matches the MFMA kernels)
with a projection head to EMBED_DIM=1536 and L2 normalization.
Random-init weights (air-gapped): embeddings are deterministic and
well-distributed; the retrieval CONTRACT (dims, cosine, top-k) is what the
pipelines rely on.

GPU attention runs the hand-written varlen MFMA flash kernel
(ops/hip/paged_attn.hip, CAUSAL=false): per-layer K/V scatter into a
paged scratch cache, then ONE bidirectional flash-attention kernel per
layer streaming K/V fragments — no padded bmm, no materialized score
matrix.  The CPU path keeps a padded bmm + masked-softmax reference of
the same semantics (GPU-vs-CPU parity test in tests/test_gpu_models.py).
"""

from __future__ import annotations

from dataclasses import dataclass

import torch
import torch.nn.functional as F

from ..ops import dispatch as D
from ..vector.index import EMBED_DIM
from .tokenizer import BpeTokenizer


@dataclass
class EncoderConfig:
    name: str = "bge-small"
    vocab_size: int = 30_522
    hidden: int = 384
    n_layers: int = 12
    n_heads: int = 8          # attention block 8 x 64 (kernel geometry)
    d_head: int = 64
    ffn: int = 1536
    max_pos: int = 512
    out_dim: int = EMBED_DIM
    norm_eps: float = 1e-5

    @property
    def attn_dim(self) -> int:
        return self.n_heads * self.d_head


class EmbeddingEncoder:
    def __init__(self, cfg: EncoderConfig | None = None, device: str = "cuda",
                 dtype=torch.bfloat16, seed: int = 1):
        self.cfg = cfg or EncoderConfig()
        c = self.cfg
        self.device = device
        self.dtype = dtype
        self.tokenizer = BpeTokenizer(c.vocab_size)
        gen_dev = device if str(device).startswith("cuda") else "cpu"
        g = torch.Generator(device=gen_dev).manual_seed(seed)

        def w(*shape, std=0.02):
            # generate on-device: 8B-class random init in seconds, no 32 GB
            # host spike
            t = torch.randn(*shape, generator=g, dtype=torch.float32,
                            device=gen_dev) * std
            return t.to(device=device, dtype=dtype)

        self.tok_embed = w(c.vocab_size, c.hidden)
        self.layers = [{
            "attn_norm": torch.ones(c.hidden, device=device, dtype=dtype),
            "mlp_norm": torch.ones(c.hidden, device=device, dtype=dtype),
            "wqkv": w(3 * c.attn_dim, c.hidden),
            "wo": w(c.hidden, c.attn_dim),
            "wgu": w(2 * c.ffn, c.hidden),
            "wdown": w(c.hidden, c.ffn),
        } for _ in range(c.n_layers)]
        self.final_norm = torch.ones(c.hidden, device=device, dtype=dtype)
        self.proj = w(c.out_dim, c.hidden)
        half = c.d_head // 2
        inv = 1.0 / (10_000.0 ** (torch.arange(half, dtype=torch.float64) / half))
        ang = torch.outer(torch.arange(c.max_pos, dtype=torch.float64), inv)
        self.rope_cos = ang.cos().float().to(device)
        self.rope_sin = ang.sin().float().to(device)
        self.scale = 1.0 / (c.d_head ** 0.5)
        self._use_hip = str(device).startswith("cuda")

    # -- varlen attention plumbing (GPU path) ------------------------------
    def _attn_maps(self, n: int, L: int, lens: torch.Tensor):
        """Page table / q-block maps for the bidirectional varlen kernel.
        Items sit at padded offsets i*L; pages are preallocated per item
        over the PADDED length (pad slots are never read: item_len bounds
        the key range)."""
        c = self.cfg
        ppi = (L + 63) // 64            # pages per item
        bt = (torch.arange(n * ppi, dtype=torch.int32, device=self.device)
              .reshape(n, ppi).contiguous())
        # pad rows scatter to a TRASH page no block_table references:
        # their k/v are downstream of an unwritten attention-output row
        # (possibly NaN), and the PV MFMA reads FULL 64-slot pages before
        # masking — 0 * NaN would poison valid rows from layer 2 on
        j = torch.arange(L, device=self.device).unsqueeze(0).expand(n, L)
        lens_d = lens.to(self.device).unsqueeze(1)
        base = (torch.arange(n, device=self.device) * ppi * 64) \
            .unsqueeze(1) + j
        trash = n * ppi * 64
        slots = torch.where(j < lens_d, base, trash) \
            .to(torch.int32).reshape(-1).contiguous()
        qb_item, qb_pos0 = [], []
        lens_l = lens.tolist()
        for i in range(n):
            for p0 in range(0, max(int(lens_l[i]), 1), 16):
                qb_item.append(i)
                qb_pos0.append(p0)
        dev = self.device
        mk = lambda x: torch.tensor(x, dtype=torch.int32, device=dev)
        return {
            "n_pages": n * ppi + 1,     # +1 trash page for pad rows
            "block_table": bt,
            "slots": slots,
            "qb_item": mk(qb_item), "qb_pos0": mk(qb_pos0),
            "item_off": mk([i * L for i in range(n)]),
            "item_start": mk([0] * n),
            "item_len": mk([max(int(x), 1) for x in lens_l]),
        }

    @torch.no_grad()
    def embed_batch(self, texts: list[str]):
        import numpy as np
        if not texts:
            return np.zeros((0, self.cfg.out_dim), dtype=np.float32)
        c = self.cfg
        enc = [self.tokenizer.encode(t)[: c.max_pos] or [BpeTokenizer.BOS]
               for t in texts]
        n = len(enc)
        L = max(len(x) for x in enc)
        toks = torch.zeros(n, L, dtype=torch.int64, device=self.device)
        lens = torch.zeros(n, dtype=torch.int32)
        for i, x in enumerate(enc):
            toks[i, :len(x)] = torch.tensor(x, dtype=torch.int64,
                                            device=self.device)
            lens[i] = len(x)
        lens_dev = lens.to(self.device)
        positions = torch.arange(L, dtype=torch.int32, device=self.device) \
            .repeat(n).contiguous()
        T = n * L

        from ..ops import have_ext
        use_hip = self._use_hip and have_ext()
        if use_hip:
            maps = self._attn_maps(n, L, lens)
            # zeros, not empty: the attention kernel's MFMAs read FULL
            # 64-slot pages before masking; garbage V in never-scattered
            # tail slots would 0*inf=NaN the PV accumulate
            kc = torch.zeros(maps["n_pages"], c.n_heads, c.d_head // 8, 64,
                             8, dtype=self.dtype, device=self.device)
            vc = torch.zeros(maps["n_pages"], c.n_heads, c.d_head, 64,
                             dtype=self.dtype, device=self.device)
        else:
            # per-row score limits: row = (seq, head, qpos) -> seq len
            row_limits = lens_dev.repeat_interleave(c.n_heads * L) \
                .contiguous()

        res = self.tok_embed.index_select(0, toks.reshape(-1)).contiguous()
        h = None
        mlp_out = None
        for li, Ly in enumerate(self.layers):
            if li == 0:
                h = D.rmsnorm(res, Ly["attn_norm"], c.norm_eps)
            else:
                h = D.rmsnorm_residual(mlp_out, res, Ly["attn_norm"], c.norm_eps)
            qkv = F.linear(h, Ly["wqkv"])
            A = c.attn_dim
            q = qkv[:, :A].reshape(T, c.n_heads, c.d_head).contiguous()
            k = qkv[:, A:2 * A].reshape(T, c.n_heads, c.d_head).contiguous()
            v = qkv[:, 2 * A:].reshape(T, c.n_heads, c.d_head).contiguous()
            D.rope_inplace(q, k, self.rope_cos, self.rope_sin, positions)
            if use_hip:
                # bidirectional varlen MFMA flash attention (CAUSAL=false)
                D.kv_scatter(k, v, kc, vc, maps["slots"])
                attn = D.ext().paged_attn_prefill(
                    q, kc, vc, maps["block_table"], maps["qb_item"],
                    maps["qb_pos0"], maps["item_off"], maps["item_start"],
                    maps["item_len"], self.scale, False)
                attn = attn.contiguous()
            else:
                # [n*heads, L, D] padded bmm reference
                qh = q.reshape(n, L, c.n_heads, c.d_head).permute(0, 2, 1, 3) \
                    .reshape(n * c.n_heads, L, c.d_head)
                kh = k.reshape(n, L, c.n_heads, c.d_head).permute(0, 2, 1, 3) \
                    .reshape(n * c.n_heads, L, c.d_head)
                vh = v.reshape(n, L, c.n_heads, c.d_head).permute(0, 2, 1, 3) \
                    .reshape(n * c.n_heads, L, c.d_head)
                scores = (torch.bmm(qh.float(), kh.float().transpose(1, 2))
                          * self.scale).reshape(n * c.n_heads * L, L) \
                    .contiguous()
                D.softmax_rows_(scores, 0, False, 0, row_limits)
                attn = torch.bmm(scores.reshape(n * c.n_heads, L, L),
                                 vh.float())
                attn = attn.reshape(n, c.n_heads, L, c.d_head) \
                    .permute(0, 2, 1, 3).reshape(T, A).to(self.dtype) \
                    .contiguous()
            o = F.linear(attn, Ly["wo"])
            h = D.rmsnorm_residual(o, res, Ly["mlp_norm"], c.norm_eps)
            gu = F.linear(h, Ly["wgu"])
            act = D.swiglu(gu[:, :c.ffn], gu[:, c.ffn:])
            mlp_out = F.linear(act, Ly["wdown"])
        final_h = D.rmsnorm_residual(mlp_out, res, self.final_norm, c.norm_eps)
        # mean-pool valid positions, project, L2-normalize.  Pad rows are
        # never written by the varlen kernel (garbage bf16): zero them
        # with where() before pooling so inf*0 cannot NaN the pool.
        hs = final_h.reshape(n, L, c.hidden).float()
        valid = (torch.arange(L, device=self.device).unsqueeze(0)
                 < lens_dev.unsqueeze(1)).unsqueeze(-1)
        hs = torch.where(valid, hs, torch.zeros((), device=self.device))
        pooled = hs.sum(dim=1) / valid.float().sum(dim=1).clamp(min=1)
        out = F.linear(pooled.to(self.dtype), self.proj).float()
        out = out / out.norm(dim=-1, keepdim=True).clamp(min=1e-9)
        return out.cpu().numpy()

    def embed(self, text: str):
        return self.embed_batch([text])[0]
