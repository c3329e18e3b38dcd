"""Llama-family decoder for the agent LLM (K4), MI355X-native.

The reference's textgen model is a managed Bedrock/Azure endpoint
(terraform/core/main.tf:461,495); here it is an on-GPU decoder built from
the hand-written HIP kernels (rmsnorm / rope / swiglu / paged-attention
decode) plus rocBLAS GEMMs via torch.nn.functional.linear for the plain
projections.  Weights are random-init bf16 (the north-star benchmark runs
synthetic data / random weights; the architecture is the named config).

Decode path (the hot path): fully fused per layer —
  rmsnorm_residual -> qkv (weight-streaming skinny GEMM or rocBLAS by
  measured winner per shape) -> fused rope+paged-KV-append ->
  MFMA flash-decode paged attention -> o GEMM -> rmsnorm_residual ->
  gate_up GEMM -> swiglu -> down GEMM; the whole step is hipGraph-captured
  and self-feeding (models/serve.py).
Prefill: fused per-token projections over the varlen concatenation, then
ONE varlen flash-attention kernel per layer streaming K/V straight from
the paged cache (ops/hip/paged_attn.hip); the CPU test path keeps a
padded-batch bmm + masked-softmax reference of the same semantics.

Tensor parallelism (Megatron-style, RCCL over xGMI): attention heads and
FFN columns shard across the TP group; wo / wdown are row-parallel with ONE
all-reduce each per layer.  Weight init is deterministic — every rank
generates the identical full tensor from the shared seed and slices its
shard, so TP=N matches TP=1 numerics.  8-way xGMI is point-to-point
(7 links x ~153 GB/s), so decode-sized all-reduces use the default RCCL
algorithm which picks direct p2p for small messages.

Ops route through ops.dispatch: HIP kernels on cuda tensors (fail-loud if
the extension is missing), reference semantics on CPU so the engine and
the TP path run under multi-process gloo tests.
"""

from __future__ import annotations

from dataclasses import dataclass

import torch
import torch.nn.functional as F

from ..ops import dispatch as D
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
    attn_bias: bool = False     # Qwen2-family: biases on the q/k/v proj

    @classmethod
    def preset(cls, name: str) -> "LlamaConfig":
        if name in ("llama3-8b", "llama-3-8b"):
            return cls()
        if name == "llama3-1b":
            return cls(name=name, vocab_size=128_256, hidden=2048, n_layers=16,
                       n_q_heads=32, n_kv_heads=8, d_head=64, ffn=8192)
        if name == "tiny":
            return cls(name=name, vocab_size=2048, hidden=256, n_layers=2,
                       n_q_heads=4, n_kv_heads=2, d_head=64, ffn=512,
                       max_pos=2048)
        if name in ("qwen2-7b", "qwen2.5-7b"):
            # Qwen2-7B geometry: GQA 28/4, wide FFN, qkv biases,
            # rope theta 1e6 (the other mainstream dense-7B family)
            return cls(name=name, vocab_size=152_064, hidden=3584,
                       n_layers=28, n_q_heads=28, n_kv_heads=4, d_head=128,
                       ffn=18_944, rope_theta=1_000_000.0, max_pos=8192,
                       attn_bias=True)
        if name == "tiny-qwen":
            return cls(name=name, vocab_size=2048, hidden=256, n_layers=2,
                       n_q_heads=4, n_kv_heads=2, d_head=64, ffn=512,
                       max_pos=2048, rope_theta=1_000_000.0,
                       attn_bias=True)
        raise ValueError(f"unknown preset {name}")


class LlamaModel:
    def __init__(self, cfg: LlamaConfig, device: str = "cuda",
                 dtype=torch.bfloat16, seed: int = 0,
                 tp_rank: int = 0, tp_size: int = 1, tp_group=None):
        self.cfg = cfg
        self.device = device
        self.dtype = dtype
        self.tp_rank = tp_rank
        self.tp_size = tp_size
        self.tp_group = tp_group
        c = cfg
        assert c.n_q_heads % tp_size == 0, "q heads must divide TP"
        assert c.n_kv_heads % tp_size == 0, "kv heads must divide TP"
        assert c.ffn % tp_size == 0, "ffn must divide TP"
        self.n_q = c.n_q_heads // tp_size       # local q heads
        self.n_kv = c.n_kv_heads // tp_size     # local kv heads
        self.ffn_local = c.ffn // tp_size
        gen_dev = device if str(device).startswith("cuda") else "cpu"
        g = torch.Generator(device=gen_dev).manual_seed(seed)

        def w(*shape, std=0.02):
            # generate on-device: 8B-class random init in seconds, no 32 GB
            # host spike. Full tensor on every rank -> deterministic shards.
            t = torch.randn(*shape, generator=g, dtype=torch.float32,
                            device=gen_dev) * std
            return t.to(device=device, dtype=dtype)

        def rows(t: torch.Tensor, n_shards: int) -> torch.Tensor:
            """This rank's contiguous row shard (column-parallel weight)."""
            if n_shards == 1:
                return t
            sz = t.shape[0] // n_shards
            return t[self.tp_rank * sz:(self.tp_rank + 1) * sz].contiguous()

        def cols(t: torch.Tensor, n_shards: int) -> torch.Tensor:
            """This rank's contiguous col shard (row-parallel weight)."""
            if n_shards == 1:
                return t
            sz = t.shape[1] // n_shards
            return t[:, self.tp_rank * sz:(self.tp_rank + 1) * sz].contiguous()

        self.qkv_dim = (self.n_q + 2 * self.n_kv) * c.d_head
        self.embed = w(c.vocab_size, c.hidden)
        self.lm_head = w(c.vocab_size, c.hidden)
        self.final_norm = torch.ones(c.hidden, device=device, dtype=dtype)
        self.layers = []
        for _ in range(c.n_layers):
            wq_full = w(c.n_q_heads * c.d_head, c.hidden)
            wk_full = w(c.n_kv_heads * c.d_head, c.hidden)
            wv_full = w(c.n_kv_heads * c.d_head, c.hidden)
            wqkv = torch.cat([rows(wq_full, tp_size), rows(wk_full, tp_size),
                              rows(wv_full, tp_size)], dim=0).contiguous()
            wo = cols(w(c.hidden, c.n_q_heads * c.d_head), tp_size)
            layer = {
                "attn_norm": torch.ones(c.hidden, device=device, dtype=dtype),
                "mlp_norm": torch.ones(c.hidden, device=device, dtype=dtype),
                "wqkv": wqkv, "wo": wo,
            }
            if c.attn_bias:
                bq = w(c.n_q_heads * c.d_head)
                bk = w(c.n_kv_heads * c.d_head)
                bv = w(c.n_kv_heads * c.d_head)
                layer["bqkv"] = torch.cat(
                    [rows(bq, tp_size), rows(bk, tp_size),
                     rows(bv, tp_size)]).contiguous()
            layer.update(self._ffn_weights(w, rows, cols))
            self.layers.append(layer)
        # rope tables
        half = c.d_head // 2
        inv = 1.0 / (c.rope_theta **
                     (torch.arange(half, dtype=torch.float64) / half))
        ang = torch.outer(torch.arange(c.max_pos, dtype=torch.float64), inv)
        self.rope_cos = ang.cos().float().to(device)
        self.rope_sin = ang.sin().float().to(device)
        self.scale = 1.0 / (c.d_head ** 0.5)
        # decode-path weight packing (288 GB HBM affords the duplicates):
        # fp8 (e4m3 per-channel, W8A16) stream copies by default — the
        # decode projections are weight-bandwidth-bound, so fp8 halves the
        # stream (ops/hip/skinny_gemm_fp8.hip); QSA_FP8=0 falls back to
        # the bf16 fragment copies for the small shapes.
        import os as _os
        self.use_fp8 = _os.environ.get("QSA_FP8", "1") == "1"
        self.lm_head_f = None
        self.lm_head_f8 = None
        if str(device).startswith("cuda") and dtype == torch.bfloat16:
            if self.use_fp8:
                for L in self.layers:
                    for key in ("wqkv", "wo", "wgu", "wdown"):
                        wt = L.get(key)
                        if wt is not None and D.can_pack_weight(*wt.shape):
                            L[key + "_f8"] = D.pack_weight_fp8(wt)
                if D.can_pack_weight(*self.lm_head.shape):
                    self.lm_head_f8 = D.pack_weight_fp8(self.lm_head)
            else:
                def want(shape):
                    # only shapes where bf16 skinny beats rocBLAS
                    # (kernel_bench): duplicating big-N weights buys nothing
                    return D.can_pack_weight(*shape) and \
                        shape[0] * shape[1] <= 6144 * 4096
                for L in self.layers:
                    for key in ("wqkv", "wo", "wgu", "wdown"):
                        wt = L.get(key)
                        if wt is not None and want(wt.shape):
                            L[key + "_f"] = D.pack_weight_frag(wt)
                if want(self.lm_head.shape):
                    self.lm_head_f = D.pack_weight_frag(self.lm_head)

    def _ffn_weights(self, w, rows, cols) -> dict:
        """Per-layer FFN weights; the generator order is part of the
        deterministic-init contract (same seed => same full tensors on
        every rank, sliced per shard)."""
        c = self.cfg
        wgu_full = w(2 * c.ffn, c.hidden)
        wgu = torch.cat([rows(wgu_full[:c.ffn], self.tp_size),
                         rows(wgu_full[c.ffn:], self.tp_size)],
                        dim=0).contiguous()
        wdown = cols(w(c.hidden, c.ffn, std=0.02 / (2 * c.n_layers) ** 0.5),
                     self.tp_size)
        return {"wgu": wgu, "wdown": wdown}

    def new_kv_cache(self, n_pages: int) -> PagedKVCache:
        return PagedKVCache(self.cfg.n_layers, self.n_kv, self.cfg.d_head,
                            n_pages, self.device, self.dtype)

    def _linear(self, x: torch.Tensor, w: torch.Tensor,
                wf: torch.Tensor | None = None,
                wf8: tuple | None = None) -> torch.Tensor:
        """Projection: decode-sized cuda batches go through the
        weight-streaming skinny GEMM — fp8 (W8A16, half the weight
        stream) when packed, else bf16; everything else rocBLAS."""
        n, k = w.shape
        M = x.shape[0]
        if x.is_cuda and wf8 is not None:
            if M <= 32:
                return D.skinny_linear_fp8(x, wf8[0], wf8[1], n, k)
            # 32 < M <= 256 (continuous-batching decode): the batched fp8
            # kernel (ops/hip/gemm_fp8_batch.hip) is measured SLOWER than
            # rocBLAS bf16 at these M on MI355X — the problem is L2-/
            # compute-bound, not weight-stream-bound, so halved weight
            # bytes don't pay (profiles/fp8_decode_gemm.md).  Keep it
            # opt-in for further tuning; rocBLAS is the measured winner.
            import os as _os
            if _os.environ.get("QSA_FP8_BATCH") == "1" and M <= 256 \
                    and n % 64 == 0 and n <= 32768:
                wgs = n // 64
                splitk = 1 if wgs >= 256 else (2 if wgs >= 128 else 4)
                if k % (64 * splitk) == 0:
                    return D.gemm_fp8_batch(x, wf8[0], wf8[1], n, k,
                                            splitk)
        if x.is_cuda and M <= 32 and wf is not None and \
                n * k <= 6144 * 4096:
            # bf16 skinny wins only the small decode shapes (qkv/wo);
            # rocBLAS is at the bandwidth floor for the big-N ones
            return D.skinny_linear(x, wf, n, k)
        return F.linear(x, w)

    def _tp_all_reduce(self, t: torch.Tensor) -> torch.Tensor:
        if self.tp_size > 1:
            import torch.distributed as dist
            dist.all_reduce(t, group=self.tp_group)
        return t

    # ------------------------------------------------------------------
    def _split_qkv(self, qkv: torch.Tensor, n: int):
        """Strided [n, H, D] views into the fused qkv buffer — the rope /
        kv_scatter kernels take row strides, no copies."""
        c = self.cfg
        qd = self.n_q * c.d_head
        kd = self.n_kv * c.d_head
        q = qkv[:, :qd].view(n, self.n_q, c.d_head)
        k = qkv[:, qd:qd + kd].view(n, self.n_kv, c.d_head)
        v = qkv[:, qd + kd:].view(n, self.n_kv, c.d_head)
        return q, k, v

    @torch.no_grad()
    def forward_decode(self, tokens: torch.Tensor, kv: PagedKVCache,
                       block_table: torch.Tensor, seq_lens: torch.Tensor,
                       positions: torch.Tensor) -> torch.Tensor:
        """tokens [B] i64 -> logits [B, vocab].  seq_lens INCLUDE the new
        token (its k/v is appended at position seq_lens-1)."""
        c = self.cfg
        B = tokens.shape[0]
        res = self.embed.index_select(0, tokens).contiguous()
        h = None
        mlp_out = None
        for li, L in enumerate(self.layers):
            if li == 0:
                h = D.rmsnorm(res, L["attn_norm"], c.norm_eps)
            else:
                h = D.rmsnorm_residual(mlp_out, res, L["attn_norm"],
                                       c.norm_eps)
            qkv = self._linear(h, L["wqkv"], L.get("wqkv_f"),
                               L.get("wqkv_f8"))
            if "bqkv" in L:
                qkv = qkv + L["bqkv"]
            # strided [B, H, D] views straight into the fused qkv buffer —
            # the kernels take row strides, no contiguous() copies
            qd = self.n_q * c.d_head
            kd = self.n_kv * c.d_head
            q = qkv[:, :qd].view(B, self.n_q, c.d_head)
            k = qkv[:, qd:qd + kd].view(B, self.n_kv, c.d_head)
            v = qkv[:, qd + kd:].view(B, self.n_kv, c.d_head)
            D.rope_kv_append(q, k, v, kv.k[li], kv.v[li], self.rope_cos,
                             self.rope_sin, block_table, seq_lens)
            attn = D.paged_attn_decode(q, kv.k[li], kv.v[li], block_table,
                                       seq_lens, self.scale)
            o = self._tp_all_reduce(
                self._linear(attn.view(B, -1), L["wo"], L.get("wo_f"),
                             L.get("wo_f8")))
            h = D.rmsnorm_residual(o, res, L["mlp_norm"], c.norm_eps)
            mlp_out = self._ffn(L, h)
        final_h = D.rmsnorm_residual(mlp_out, res, self.final_norm, c.norm_eps)
        return self._linear(final_h, self.lm_head, self.lm_head_f,
                            self.lm_head_f8)

    def _ffn(self, L: dict, h: torch.Tensor) -> torch.Tensor:
        """Dense SwiGLU FFN (TP row/col-parallel). Mixtral overrides with
        the routed MoE (models/mixtral.py)."""
        gu = self._linear(h, L["wgu"], L.get("wgu_f"),
                          L.get("wgu_f8"))
        act = D.swiglu(gu[:, :self.ffn_local], gu[:, self.ffn_local:])
        return self._tp_all_reduce(
            self._linear(act, L["wdown"], L.get("wdown_f"),
                         L.get("wdown_f8")))

    def _gather_kv(self, kc_l: torch.Tensor, vc_l: torch.Tensor,
                   pages: list[int], n: int):
        """Gather positions [0, n) of a sequence from the paged cache ->
        (K [KVH, n, D], V [KVH, n, D]) contiguous for the prefill GEMMs."""
        c = self.cfg
        idx = torch.tensor(pages, dtype=torch.int64, device=self.device)
        kpg = kc_l.index_select(0, idx)        # [np, KVH, D/8, 64, 8]
        kseq = kpg.permute(1, 0, 3, 2, 4).reshape(
            self.n_kv, len(pages) * 64, c.d_head)[:, :n]
        vpg = vc_l.index_select(0, idx)        # [np, KVH, D, 64]
        vseq = vpg.permute(1, 0, 3, 2).reshape(
            self.n_kv, len(pages) * 64, c.d_head)[:, :n]
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

        # ---- attention prep (once per prefill call) ---------------------
        R = self.n_q // self.n_kv
        nb = len(items)
        nmax = max(lens)
        npmax = max((int(items[i][2]) + lens[i] + kv.PAGE - 1) // kv.PAGE
                    for i in range(nb))
        ctxp = npmax * kv.PAGE
        import os as _os
        use_hip_prefill = str(dev).startswith("cuda") and \
            self.cfg.d_head in (64, 128) and self.n_q % 4 == 0 and \
            _os.environ.get("QSA_NO_HIP_PREFILL") != "1"
        # optional token-dim bucketing (QSA_PREFILL_PAD=1): pad the
        # concatenated batch to a bucket with TunableOp-tuned GEMM
        # entries (data/tunableop_mi355x.csv).  Pad rows run the
        # projections but are NEVER scattered into the KV cache (the
        # scatter uses sliced [:T] views) and no attention q-block maps
        # cover them, so their garbage stays confined row-wise.
        T_pad = T
        if use_hip_prefill and _os.environ.get("QSA_PREFILL_PAD") == "1":
            for b in (4096, 8192, 16384, 24576, 32768, 40960, 49152,
                      57344, 65536):
                if T <= b:
                    T_pad = b
                    break
            else:
                T_pad = T
            if T_pad > T:
                padn = T_pad - T
                tokens = torch.cat([
                    tokens, torch.zeros(padn, dtype=torch.int64,
                                        device=dev)])
                positions = torch.cat([
                    positions,
                    torch.zeros(padn, dtype=torch.int32, device=dev)
                ]).contiguous()
        if use_hip_prefill:
            # varlen flash prefill over the paged cache: per-q-block maps
            # (one WAVE per 16 q rows x head in the kernel)
            qb_item, qb_pos0 = [], []
            for i, n in enumerate(lens):
                for p0 in range(0, n, 16):
                    qb_item.append(i)
                    qb_pos0.append(p0)
            qb_item_t = torch.tensor(qb_item, dtype=torch.int32, device=dev)
            qb_pos0_t = torch.tensor(qb_pos0, dtype=torch.int32, device=dev)
            off_t = torch.tensor(offs[:-1], dtype=torch.int32, device=dev)
            start_t = torch.tensor([int(s) for _, _, s in items],
                                   dtype=torch.int32, device=dev)
            len_t = torch.tensor(lens, dtype=torch.int32, device=dev)
            bt = torch.zeros(nb, npmax, dtype=torch.int32)
            for i, (_, sid, _) in enumerate(items):
                pages = kv._seq_pages[sid]
                bt[i, :len(pages)] = torch.tensor(pages, dtype=torch.int32)
            bt_t = bt.to(dev)
        else:
            # CPU reference: padded-batch bmm pair + masked softmax.
            # Per-row limits (start_i + qpos + 1, or 0 on pad rows) give
            # exact causal + ragged masking.
            dst_idx = torch.tensor(
                [i * nmax + p for i, n in enumerate(lens) for p in range(n)],
                dtype=torch.int64, device=dev)
            page_idx = torch.zeros(nb, npmax, dtype=torch.int64)
            for i, (_, sid, _) in enumerate(items):
                pages = kv._seq_pages[sid]
                page_idx[i, :len(pages)] = torch.tensor(pages,
                                                        dtype=torch.int64)
            flat_pages = page_idx.reshape(-1).to(dev)
            limits = torch.zeros(nb, self.n_q, nmax, dtype=torch.int32)
            for i, ((_, _, start), n) in enumerate(zip(items, lens)):
                lim = torch.arange(start + 1, start + n + 1,
                                   dtype=torch.int32)
                limits[i, :, :n] = lim.unsqueeze(0)
            row_limits = limits.reshape(-1).to(dev)

        res = self.embed.index_select(0, tokens).contiguous()
        h = None
        mlp_out = None
        for li, L in enumerate(self.layers):
            if li == 0:
                h = D.rmsnorm(res, L["attn_norm"], c.norm_eps)
            else:
                h = D.rmsnorm_residual(mlp_out, res, L["attn_norm"],
                                       c.norm_eps)
            qkv = F.linear(h, L["wqkv"], L.get("bqkv"))
            q, k, v = self._split_qkv(qkv, T_pad)
            D.rope_inplace(q, k, self.rope_cos, self.rope_sin, positions)
            D.kv_scatter(k[:T], v[:T], kv.k[li], kv.v[li], slots)
            if use_hip_prefill:
                # ONE varlen flash kernel: streams K/V pages directly, no
                # gather/pad/score materialization
                attn = D.ext().paged_attn_prefill(
                    q, kv.k[li], kv.v[li], bt_t, qb_item_t, qb_pos0_t,
                    off_t, start_t, len_t, self.scale, True)
            else:
                # padded q: [nb*nmax, n_q, D] -> [nb*KVH, R*nmax, D]
                q_pad = torch.zeros(nb * nmax, self.n_q, c.d_head,
                                    dtype=self.dtype, device=dev)
                q_pad.index_copy_(0, dst_idx, q)
                qb = q_pad.view(nb, nmax, self.n_kv, R, c.d_head) \
                    .permute(0, 2, 3, 1, 4) \
                    .reshape(nb * self.n_kv, R * nmax, c.d_head)
                # gather padded K/V from the paged cache (just scattered)
                ksel = kv.k[li].index_select(0, flat_pages) \
                    .view(nb, npmax, self.n_kv, c.d_head // 8, kv.PAGE, 8) \
                    .permute(0, 2, 1, 4, 3, 5) \
                    .reshape(nb * self.n_kv, ctxp, c.d_head)
                vsel = kv.v[li].index_select(0, flat_pages) \
                    .view(nb, npmax, self.n_kv, c.d_head, kv.PAGE) \
                    .permute(0, 2, 1, 4, 3) \
                    .reshape(nb * self.n_kv, ctxp, c.d_head)
                # bf16 scores + fused-scale masked softmax
                scores = torch.bmm(qb, ksel.transpose(1, 2)) \
                    .reshape(-1, ctxp)
                D.softmax_rows_bf16_(scores, self.scale, row_limits)
                probs = scores.reshape(nb * self.n_kv, R * nmax, ctxp)
                a = torch.bmm(probs, vsel)
                attn = a.view(nb, self.n_kv, R, nmax, c.d_head) \
                    .permute(0, 3, 1, 2, 4) \
                    .reshape(nb * nmax, self.n_q * c.d_head) \
                    .index_select(0, dst_idx)
            o = self._tp_all_reduce(F.linear(attn, L["wo"]))
            h = D.rmsnorm_residual(o, res, L["mlp_norm"], c.norm_eps)
            mlp_out = self._ffn(L, h)
        final_h = D.rmsnorm_residual(mlp_out, res, self.final_norm, c.norm_eps)
        last = torch.tensor([offs[i] + lens[i] - 1 for i in range(len(items))],
                            dtype=torch.int64, device=dev)
        return self._linear(final_h.index_select(0, last), self.lm_head,
                            self.lm_head_f, self.lm_head_f8)

    @torch.no_grad()
    def forward_prefill(self, tokens: torch.Tensor, kv: PagedKVCache,
                        seq_id: int) -> torch.Tensor:
        """Single-sequence prefill (wrapper over the batch path)."""
        return self.forward_prefill_batch([(tokens, seq_id, 0)], kv)[0]
