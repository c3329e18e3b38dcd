#!/usr/bin/env python3
"""Minimal dispatch set for PMC counter collection (a handful of launches
of the two flagship kernels only)."""
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch

from quickstart_streaming_agents_amd.models.kv_cache import PagedKVCache
from quickstart_streaming_agents_amd.ops import ext

e = ext()
dev = "cuda:0"
M, N, K = 24, 28672, 4096
a = torch.randn(M, K, device=dev, dtype=torch.bfloat16) * 0.1
w = torch.randn(N, K, device=dev, dtype=torch.bfloat16) * 0.02
wf = e.pack_weight_frag(w)
for _ in range(3):
    out = e.skinny_gemm(a, wf, N, K)

B, QH, KVH, D, ctx = 128, 32, 8, 128, 512
kv = PagedKVCache(1, KVH, D, 2048, device=dev)
for s in range(B):
    kv.allocate(s, ctx)
bt = kv.block_table(list(range(B)))
sl = kv.seq_lens_tensor(list(range(B)))
kv.k[0].normal_()
kv.v[0].normal_()
q = torch.randn(B, QH, D, device=dev, dtype=torch.bfloat16)
for _ in range(3):
    o = e.paged_attn_decode(q, kv.k[0], kv.v[0], bt, sl, 0.088)
torch.cuda.synchronize()
print("pmc probe done")
