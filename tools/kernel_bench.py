#!/usr/bin/env python3
"""Isolated kernel A/B: skinny_gemm vs rocBLAS (F.linear) per decode shape,
and paged-attention decode at bench geometry.  Within-process interleaved
timing (guide §5.4 rule 24)."""

import argparse
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch  # noqa: E402
import torch.nn.functional as F  # noqa: E402


def t_ms(fn, iters=50, warmup=10):
    for _ in range(warmup):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters * 1e3


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--batch", type=int, default=24)
    p.add_argument("--ctx", type=int, default=512)
    args = p.parse_args()
    assert torch.cuda.is_available()
    from quickstart_streaming_agents_amd.ops import ext
    e = ext()
    M = args.batch
    dev = "cuda:0"
    shapes = [("qkv", 6144, 4096), ("wo", 4096, 4096),
              ("wgu", 28672, 4096), ("wdown", 4096, 14336),
              ("lm_head", 128256, 4096)]
    print(f"M={M}  (floor = W bytes / 6.3 TB/s)")
    for name, N, K in shapes:
        a = torch.randn(M, K, device=dev, dtype=torch.bfloat16) * 0.1
        w = torch.randn(N, K, device=dev, dtype=torch.bfloat16) * 0.02
        wf = e.pack_weight_frag(w)
        ms_blas = t_ms(lambda: F.linear(a, w))
        if M <= 32:
            ms_sk = t_ms(lambda: e.skinny_gemm(a, wf, N, K))
        else:
            ms_sk = float("nan")  # custom path is M<=32; rocBLAS above
        floor = N * K * 2 / 6.3e12 * 1e3
        print(f"{name:8s} N={N:6d} K={K:6d}: rocblas {ms_blas*1e3:8.1f} us  "
              f"skinny {ms_sk*1e3:8.1f} us  floor {floor*1e3:8.1f} us  "
              f"(skinny {floor/ms_sk*100:.0f}% SOL)")
        if os.environ.get("QSA_SWEEP") == "1":
            for waves, nt, var, tag in [
                    (4, 1, 0, "w4 nt  full"), (4, 0, 0, "w4 pln full"),
                    (8, 1, 0, "w8 nt  full"), (8, 0, 0, "w8 pln full"),
                    (2, 1, 0, "w2 nt  full"), (1, 1, 0, "w1 nt  full"),
                    (4, 1, 1, "w4 nt  Wonly"), (4, 1, 2, "w4 nt  Aonly"),
                    (8, 1, 1, "w8 nt  Wonly")]:
                ms = t_ms(lambda: e.skinny_gemm_probe(a, wf, N, K, waves,
                                                      nt, var))
                print(f"    {tag:14s} {ms*1e3:8.1f} us ({floor/ms*100:.0f}% SOL)")
        del a, w, wf

    # paged attention at bench geometry
    from quickstart_streaming_agents_amd.models.kv_cache import PagedKVCache
    QH, KVH, D = 32, 8, 128
    kv = PagedKVCache(1, KVH, D, 1024, device=dev)
    for s in range(M):
        kv.allocate(s, args.ctx)
    bt = kv.block_table(list(range(M)))
    sl = kv.seq_lens_tensor(list(range(M)))
    kv.k[0].normal_()
    kv.v[0].normal_()
    q = torch.randn(M, QH, D, device=dev, dtype=torch.bfloat16)
    ms = t_ms(lambda: e.paged_attn_decode(q, kv.k[0], kv.v[0], bt, sl,
                                          0.088), iters=100)
    kv_bytes = M * KVH * args.ctx * D * 2 * 2
    floor = kv_bytes / 6.3e12 * 1e3
    print(f"paged_attn B={M} ctx={args.ctx}: {ms*1e3:8.1f} us  "
          f"floor {floor*1e3:8.1f} us ({floor/ms*100:.0f}% SOL, "
          f"{kv_bytes/1e6:.0f} MB KV)")


if __name__ == "__main__":
    main()
