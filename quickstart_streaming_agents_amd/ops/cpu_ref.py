"""Plain PyTorch fp32 references for every HIP kernel (numerics contracts).

GPU tests (tests/test_gpu_kernels.py) compare each qsa_hip kernel against
these on the same random inputs; tolerances account for bf16 I/O.
"""

from __future__ import annotations

import torch


def rmsnorm_ref(x: torch.Tensor, w: torch.Tensor, eps: float = 1e-5,
                residual: torch.Tensor | None = None):
    xf = x.float()
    res_out = None
    if residual is not None:
        xf = xf + residual.float()
        res_out = xf
    var = xf.pow(2).mean(dim=-1, keepdim=True)
    y = xf * torch.rsqrt(var + eps) * w.float()
    return (y, res_out) if residual is not None else y


def swiglu_ref(gate: torch.Tensor, up: torch.Tensor):
    g = gate.float()
    return torch.nn.functional.silu(g) * up.float()


def rope_tables(max_pos: int, d_head: int, theta: float = 500000.0):
    half = d_head // 2
    inv = 1.0 / (theta ** (torch.arange(half, dtype=torch.float64) / half))
    pos = torch.arange(max_pos, dtype=torch.float64)
    ang = torch.outer(pos, inv)
    return ang.cos().float(), ang.sin().float()


def rope_ref(x: torch.Tensor, pos: torch.Tensor, cos_t: torch.Tensor,
             sin_t: torch.Tensor):
    """x [B, H, D]; llama half-rotation (x0, x1=x[d+half])."""
    B, H, D = x.shape
    half = D // 2
    xf = x.float()
    c = cos_t[pos.long()].unsqueeze(1)  # [B, 1, half]
    s = sin_t[pos.long()].unsqueeze(1)
    x0, x1 = xf[..., :half], xf[..., half:]
    return torch.cat([x0 * c - x1 * s, x0 * s + x1 * c], dim=-1)


def paged_attn_ref(q: torch.Tensor, kc: torch.Tensor, vc: torch.Tensor,
                   block_table: torch.Tensor, seq_lens: torch.Tensor,
                   scale: float) -> torch.Tensor:
    """q [B, QH, D]; kc [P, KVH, D/8, 64, 8]; vc [P, KVH, D, 64]."""
    B, QH, D = q.shape
    KVH = kc.shape[1]
    R = QH // KVH
    out = torch.zeros_like(q, dtype=torch.float32)
    # un-page K into [P, KVH, 64, D]
    kn = kc.permute(0, 1, 3, 2, 4).reshape(kc.shape[0], KVH, 64, D).float()
    vn = vc.permute(0, 1, 3, 2).float()   # -> [P, KVH, 64, D]
    for b in range(B):
        n = int(seq_lens[b])
        if n == 0:
            continue
        npages = (n + 63) // 64
        pages = block_table[b, :npages].long()
        k_seq = kn[pages].permute(1, 0, 2, 3).reshape(KVH, npages * 64, D)[:, :n]
        v_seq = vn[pages].permute(1, 0, 2, 3).reshape(KVH, npages * 64, D)[:, :n]
        for h in range(QH):
            kvh = h // R
            sc = (k_seq[kvh] @ q[b, h].float()) * scale
            p = torch.softmax(sc, dim=-1)
            out[b, h] = p @ v_seq[kvh]
    return out


def topk_cosine_ref(queries: torch.Tensor, docs: torch.Tensor, k: int):
    scores = queries.float() @ docs.float().T
    s, i = torch.topk(scores, k, dim=-1)
    return s, i.int()


def softmax_rows_ref(scores: torch.Tensor, col_offset: int = 0,
                     causal: bool = False):
    s = scores.float().clone()
    rows, cols = s.shape
    if causal:
        col = torch.arange(cols, device=s.device).unsqueeze(0)
        row = torch.arange(rows, device=s.device).unsqueeze(1)
        s.masked_fill_(col > row + col_offset, float("-inf"))
    out = torch.softmax(s, dim=-1)
    return torch.nan_to_num(out, nan=0.0)


def window_agg_ref(ts: torch.Tensor, key: torch.Tensor, value, t0: int,
                   win_ms: int, nwin: int, nkeys: int):
    counts = torch.zeros(nkeys, nwin, dtype=torch.int32)
    sums = torch.zeros(nkeys, nwin, dtype=torch.float32)
    w = ((ts - t0) // win_ms)
    ok = (w >= 0) & (w < nwin)
    for i in torch.nonzero(ok).flatten().tolist():
        counts[key[i], w[i]] += 1
        if value is not None:
            sums[key[i], w[i]] += float(value[i])
    return counts, sums
