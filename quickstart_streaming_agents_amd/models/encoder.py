"""On-GPU embedding encoder (K1): bge-small-class transformer -> 1536-d.

Replaces the managed embedding models (titan-embed-v1 / ada-002, both
1536-d — terraform/core/main.tf:529,563; dims contract validate.py:56-62).
A bge-small-class bidirectional encoder (12 layers, 384 hidden) with a
projection head to EMBED_DIM=1536 and L2 normalization, built on the same
HIP kernels (rmsnorm / rope / swiglu / masked row-softmax) + rocBLAS GEMMs.
Random-init weights (air-gapped): embeddings are deterministic and
well-distributed; the retrieval CONTRACT (dims, cosine, top-k) is what the
pipelines rely on.
"""

from __future__ import annotations

from dataclasses import dataclass

import torch
import torch.nn.functional as F

from ..ops import dispatch as D
from ..vector.index import EMBED_DIM
from .tokenizer import HashTokenizer


@dataclass
class EncoderConfig:
    name: str = "bge-small"
    vocab_size: int = 30_522
    hidden: int = 384
    n_layers: int = 12
    n_heads: int = 6
    d_head: int = 64
    ffn: int = 1536
    max_pos: int = 512
    out_dim: int = EMBED_DIM
    norm_eps: float = 1e-5


class EmbeddingEncoder:
    def __init__(self, cfg: EncoderConfig | None = None, device: str = "cuda",
                 dtype=torch.bfloat16, seed: int = 1):
        self.cfg = cfg or EncoderConfig()
        c = self.cfg
        assert c.n_heads * c.d_head == c.hidden
        self.device = device
        self.dtype = dtype
        self.tokenizer = HashTokenizer(c.vocab_size)
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
            "wqkv": w(3 * c.hidden, c.hidden),
            "wo": w(c.hidden, c.hidden),
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

    @torch.no_grad()
    def embed_batch(self, texts: list[str]):
        import numpy as np
        if not texts:
            return np.zeros((0, self.cfg.out_dim), dtype=np.float32)
        c = self.cfg
        enc = [self.tokenizer.encode(t)[: c.max_pos] or [HashTokenizer.BOS]
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
        # per-row score limits: row = (seq, head, qpos) -> seq len
        row_limits = lens_dev.repeat_interleave(c.n_heads * L).contiguous()
        positions = torch.arange(L, dtype=torch.int32, device=self.device) \
            .repeat(n).contiguous()

        res = self.tok_embed.index_select(0, toks.reshape(-1)).contiguous()
        h = None
        mlp_out = None
        T = n * L
        for li, Ly in enumerate(self.layers):
            if li == 0:
                h = D.rmsnorm(res, Ly["attn_norm"], c.norm_eps)
            else:
                h = D.rmsnorm_residual(mlp_out, res, Ly["attn_norm"], c.norm_eps)
            qkv = F.linear(h, Ly["wqkv"])
            q = qkv[:, :c.hidden].reshape(T, c.n_heads, c.d_head).contiguous()
            k = qkv[:, c.hidden:2 * c.hidden].reshape(T, c.n_heads,
                                                      c.d_head).contiguous()
            v = qkv[:, 2 * c.hidden:].reshape(T, c.n_heads, c.d_head).contiguous()
            D.rope_inplace(q, k, self.rope_cos, self.rope_sin, positions)
            # [n*heads, L, D]
            qh = q.reshape(n, L, c.n_heads, c.d_head).permute(0, 2, 1, 3) \
                .reshape(n * c.n_heads, L, c.d_head)
            kh = k.reshape(n, L, c.n_heads, c.d_head).permute(0, 2, 1, 3) \
                .reshape(n * c.n_heads, L, c.d_head)
            vh = v.reshape(n, L, c.n_heads, c.d_head).permute(0, 2, 1, 3) \
                .reshape(n * c.n_heads, L, c.d_head)
            scores = (torch.bmm(qh.float(), kh.float().transpose(1, 2))
                      * self.scale).reshape(n * c.n_heads * L, L).contiguous()
            D.softmax_rows_(scores, 0, False, 0, row_limits)
            attn = torch.bmm(scores.reshape(n * c.n_heads, L, L), vh.float())
            attn = attn.reshape(n, c.n_heads, L, c.d_head).permute(0, 2, 1, 3) \
                .reshape(T, c.hidden).to(self.dtype).contiguous()
            o = F.linear(attn, Ly["wo"])
            h = D.rmsnorm_residual(o, res, Ly["mlp_norm"], c.norm_eps)
            gu = F.linear(h, Ly["wgu"])
            act = D.swiglu(gu[:, :c.ffn], gu[:, c.ffn:])
            mlp_out = F.linear(act, Ly["wdown"])
        final_h = D.rmsnorm_residual(mlp_out, res, self.final_norm, c.norm_eps)
        # mean-pool valid positions, project, L2-normalize
        hs = final_h.reshape(n, L, c.hidden).float()
        mask = (torch.arange(L, device=self.device).unsqueeze(0)
                < lens_dev.unsqueeze(1)).float().unsqueeze(-1)
        pooled = (hs * mask).sum(dim=1) / mask.sum(dim=1).clamp(min=1)
        out = F.linear(pooled.to(self.dtype), self.proj).float()
        out = out / out.norm(dim=-1, keepdim=True).clamp(min=1e-9)
        return out.cpu().numpy()

    def embed(self, text: str):
        return self.embed_batch([text])[0]
