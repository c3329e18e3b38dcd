"""Device dispatch for the op layer.

Same call surface as the qsa_hip extension (ops/hip/ops.cpp). CUDA tensors
go to the hand-written HIP/CDNA4 kernels — fail-loud if the extension is
missing (ops/__init__.py).  CPU tensors run reference implementations with
IDENTICAL semantics (including in-place mutation contracts), which lets the
engine / model / parallel layers run under multi-process gloo tests in
CPU-only CI.  A CUDA tensor NEVER falls back to the CPU path.
"""

from __future__ import annotations

import torch

from . import ext


def _cuda(t: torch.Tensor) -> bool:
    return t.is_cuda


# ---- elementwise ----------------------------------------------------------

def rmsnorm(x: torch.Tensor, w: torch.Tensor, eps: float) -> torch.Tensor:
    if _cuda(x):
        return ext().rmsnorm(x, w, eps)
    xf = x.float()
    y = xf * torch.rsqrt(xf.pow(2).mean(-1, keepdim=True) + eps) * w.float()
    return y.to(x.dtype)


def rmsnorm_residual(x: torch.Tensor, res: torch.Tensor, w: torch.Tensor,
                     eps: float) -> torch.Tensor:
    """res <- res + x (in place); returns rmsnorm(res) * w."""
    if _cuda(x):
        return ext().rmsnorm_residual(x, res, w, eps)
    res.add_(x)
    rf = res.float()
    y = rf * torch.rsqrt(rf.pow(2).mean(-1, keepdim=True) + eps) * w.float()
    return y.to(x.dtype)


def swiglu(gate: torch.Tensor, up: torch.Tensor) -> torch.Tensor:
    if _cuda(gate):
        return ext().swiglu(gate, up)
    return (torch.nn.functional.silu(gate.float()) * up.float()).to(gate.dtype)


def rope_inplace(q: torch.Tensor, k: torch.Tensor, cos_t: torch.Tensor,
                 sin_t: torch.Tensor, pos: torch.Tensor) -> None:
    if _cuda(q):
        ext().rope_inplace(q, k, cos_t, sin_t, pos)
        return
    for t in (q, k):
        D = t.shape[-1]
        half = D // 2
        c = cos_t[pos.long()].unsqueeze(1)
        s = sin_t[pos.long()].unsqueeze(1)
        tf = t.float()
        x0, x1 = tf[..., :half], tf[..., half:]
        t.copy_(torch.cat([x0 * c - x1 * s, x0 * s + x1 * c],
                          dim=-1).to(t.dtype))


def softmax_rows_(scores: torch.Tensor, col_offset: int = 0,
                  causal: bool = False, row_mod: int = 0,
                  row_limits: torch.Tensor | None = None) -> None:
    if _cuda(scores):
        ext().softmax_rows_(scores, col_offset, causal, row_mod, row_limits)
        return
    rows, cols = scores.shape
    limit = torch.full((rows,), cols, dtype=torch.int64)
    if causal:
        r = torch.arange(rows)
        pos = r % row_mod if row_mod > 0 else r
        limit = torch.minimum(limit, pos + col_offset + 1)
    if row_limits is not None:
        limit = torch.minimum(limit, row_limits.long())
    col = torch.arange(cols).unsqueeze(0)
    masked = scores.masked_fill(col >= limit.unsqueeze(1), float("-inf"))
    out = torch.softmax(masked, dim=-1)
    scores.copy_(torch.nan_to_num(out, nan=0.0))


def softmax_rows_bf16_(scores: torch.Tensor, scale: float,
                       row_limits: torch.Tensor) -> None:
    """In-place bf16 masked softmax with folded scale; beyond-limit
    columns zeroed (prefill padded-batch attention)."""
    if _cuda(scores):
        ext().softmax_rows_bf16_(scores, scale, row_limits)
        return
    rows, cols = scores.shape
    s = scores.float() * scale
    col = torch.arange(cols).unsqueeze(0)
    masked = s.masked_fill(col >= row_limits.long().unsqueeze(1),
                           float("-inf"))
    out = torch.nan_to_num(torch.softmax(masked, dim=-1), nan=0.0)
    scores.copy_(out.to(scores.dtype))


# ---- paged KV -------------------------------------------------------------

def kv_append(knew: torch.Tensor, vnew: torch.Tensor, kc: torch.Tensor,
              vc: torch.Tensor, block_table: torch.Tensor,
              seq_lens: torch.Tensor) -> None:
    """Write row b's k/v at position seq_lens[b]-1 of its paged sequence."""
    if _cuda(knew):
        ext().kv_append(knew, vnew, kc, vc, block_table, seq_lens)
        return
    B, KVH, D = knew.shape
    for b in range(B):
        n = int(seq_lens[b])
        if n <= 0:
            continue
        page = int(block_table[b, (n - 1) // 64])
        off = (n - 1) % 64
        # K layout [P, KVH, D/8, 64, 8]; V layout [P, KVH, D, 64]
        kc[page, :, :, off, :] = knew[b].reshape(KVH, D // 8, 8)
        vc[page, :, :, off] = vnew[b]


def rope_kv_append(q, k, v, kc, vc, cos_t, sin_t, block_table,
                   seq_lens) -> None:
    """Fused decode-step rope(q,k in place semantics) + cache append.
    The fused kernel rotates q in place and writes rotated k (+v) straight
    to the paged cache; CPU reference composes the two reference ops."""
    if _cuda(q):
        ext().rope_kv_append(q, k, v, kc, vc, cos_t, sin_t, block_table,
                             seq_lens)
        return
    # positions = seq_lens - 1 (clamped)
    pos = (seq_lens.long() - 1).clamp(min=0).int()
    rope_inplace(q, k, cos_t, sin_t, pos)
    kv_append(k, v, kc, vc, block_table, seq_lens)


def kv_scatter(knew: torch.Tensor, vnew: torch.Tensor, kc: torch.Tensor,
               vc: torch.Tensor, slots: torch.Tensor) -> None:
    """Write row t at global slot ids (page*64 + offset)."""
    if _cuda(knew):
        ext().kv_scatter(knew, vnew, kc, vc, slots)
        return
    T, KVH, D = knew.shape
    for t in range(T):
        s = int(slots[t])
        page, off = s // 64, s % 64
        kc[page, :, :, off, :] = knew[t].reshape(KVH, D // 8, 8)
        vc[page, :, :, off] = vnew[t]


def paged_attn_decode(q: torch.Tensor, kc: torch.Tensor, vc: torch.Tensor,
                      block_table: torch.Tensor, seq_lens: torch.Tensor,
                      scale: float) -> torch.Tensor:
    if _cuda(q):
        return ext().paged_attn_decode(q, kc, vc, block_table, seq_lens,
                                       scale)
    from .cpu_ref import paged_attn_ref
    return paged_attn_ref(q, kc, vc, block_table, seq_lens,
                          scale).to(q.dtype)


# ---- skinny decode GEMM ---------------------------------------------------

def can_pack_weight(n: int, k: int) -> bool:
    return n % 16 == 0 and k % 256 == 0


def pack_weight_frag(w: torch.Tensor) -> torch.Tensor:
    """[N,K] bf16 -> fragment-major stream layout for skinny_linear (GPU)."""
    if w.is_cuda:
        return ext().pack_weight_frag(w)
    n, k = w.shape
    return w.reshape(n // 16, 16, k // 32, 32).permute(0, 2, 1, 3).contiguous()


def skinny_linear(x: torch.Tensor, wf: torch.Tensor, n: int,
                  k: int) -> torch.Tensor:
    """x[M,K] @ W^T via the weight-streaming decode kernel (M <= 32)."""
    if x.is_cuda:
        return ext().skinny_gemm(x, wf, n, k)
    w = wf.reshape(n // 16, k // 32, 16, 32).permute(0, 2, 1, 3) \
        .reshape(n, k)
    return (x.float() @ w.float().T).to(x.dtype)


# ---- fp8 weight-only decode GEMM (W8A16, per-channel scales) -------------

def pack_weight_fp8(w: torch.Tensor) -> tuple[torch.Tensor, torch.Tensor]:
    """[N,K] bf16 -> (fp8-e4m3 fragment-pair-major stream as uint8,
    per-channel f32 scales).  Layout doc: ops/hip/skinny_gemm_fp8.hip."""
    n, k = w.shape
    assert n % 16 == 0 and k % 256 == 0
    wf = w.float()
    s = wf.abs().amax(dim=1).clamp(min=1e-12) / 448.0
    q = (wf / s[:, None]).clamp(-448.0, 448.0).to(torch.float8_e4m3fn)
    qf = q.view(torch.uint8) \
        .reshape(n // 16, 16, k // 64, 2, 4, 8) \
        .permute(0, 2, 1, 4, 3, 5).contiguous()
    return qf, s.float().contiguous()


def unpack_weight_fp8(qf: torch.Tensor, scale: torch.Tensor, n: int,
                      k: int) -> torch.Tensor:
    """Inverse of pack_weight_fp8 -> dequantized f32 [N,K] (reference)."""
    q = qf.reshape(n // 16, k // 64, 16, 4, 2, 8) \
        .permute(0, 2, 1, 4, 3, 5).reshape(n, k)
    return q.view(torch.float8_e4m3fn).float() * scale[:, None].float()


def gemm_fp8_batch(x: torch.Tensor, qf: torch.Tensor,
                   scale: torch.Tensor, n: int, k: int,
                   splitk: int = 1) -> torch.Tensor:
    """Batched-M (<=256) fp8 weight-stream GEMM (decode batches)."""
    if x.is_cuda:
        return ext().gemm_fp8_batch(x, qf, scale, n, k, splitk)
    w = unpack_weight_fp8(qf, scale, n, k)
    return (x.float() @ w.T).to(x.dtype)


def skinny_linear_fp8(x: torch.Tensor, qf: torch.Tensor,
                      scale: torch.Tensor, n: int, k: int) -> torch.Tensor:
    """x[M,K] bf16 @ dequant(Q)^T via the fp8 weight-streaming kernel."""
    if x.is_cuda:
        return ext().skinny_gemm_fp8(x, qf, scale, n, k)
    w = unpack_weight_fp8(qf, scale, n, k)
    return (x.float() @ w.T).to(x.dtype)


# ---- retrieval / streaming ------------------------------------------------

def topk_cosine(queries: torch.Tensor, docs: torch.Tensor, k: int):
    if _cuda(queries):
        return ext().topk_cosine(queries, docs, k)
    from .cpu_ref import topk_cosine_ref
    return topk_cosine_ref(queries, docs, k)


def window_agg(ts: torch.Tensor, key: torch.Tensor,
               value: torch.Tensor | None, t0: int, win_ms: int, nwin: int,
               nkeys: int):
    if _cuda(ts):
        return ext().window_agg(ts, key, value, t0, win_ms, nwin, nkeys)
    from .cpu_ref import window_agg_ref
    return window_agg_ref(ts, key, value, t0, win_ms, nwin, nkeys)


# ---- GPU hash join (K8) ---------------------------------------------------

def hash_build(keys: torch.Tensor, ts: torch.Tensor):
    """(i64 keys, i64 event ts) -> opaque latest-per-key table.
    GPU: (tkeys, tpay) HBM tensors; CPU: a dict of key -> (ts, row).
    Event times clamp to [0, 2^40): the GPU payload packs (ts << 24 |
    row), so negative or far-future timestamps would corrupt ordering —
    clamping keeps CPU/GPU semantics identical."""
    ts = ts.clamp(min=0, max=(1 << 40) - 1)
    if _cuda(keys):
        return tuple(ext().hash_build(keys, ts.contiguous()))
    table: dict = {}
    for i, (k, t) in enumerate(zip(keys.tolist(), ts.tolist())):
        prev = table.get(k)
        if prev is None or (t, i) >= prev:
            table[k] = (t, i)
    return table


def hash_probe(table, keys: torch.Tensor, min_ts: int) -> torch.Tensor:
    """Probe -> i32 row indices (-1 = absent or event ts < min_ts)."""
    if isinstance(table, tuple):
        return ext().hash_probe(table[0], table[1], keys, int(min_ts))
    out = torch.full((keys.numel(),), -1, dtype=torch.int32)
    for i, k in enumerate(keys.tolist()):
        hit = table.get(k)
        if hit is not None and hit[0] >= min_ts:
            out[i] = hit[1]
    return out
