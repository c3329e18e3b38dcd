#!/usr/bin/env python3
"""Measure the fp8 vs bf16 decode-GEMM delta on the llama3-8b shapes.

Run on the GPU box:  python tools/fp8_bench.py
Writes a markdown table to stdout (redirect into profiles/)."""

import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch  # noqa: E402

from quickstart_streaming_agents_amd.ops import dispatch as D  # noqa: E402
from quickstart_streaming_agents_amd.ops import ext  # noqa: E402

SHAPES = [  # (name, N, K) for llama3-8b decode projections
    ("qkv", 6144, 4096),
    ("wo", 4096, 4096),
    ("wgu", 28672, 4096),
    ("wdown", 4096, 14336),
    ("lm_head", 128256, 4096),
]
M = 24
REPS = 50


def timeit(fn, reps=REPS):
    for _ in range(5):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(reps):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / reps * 1e6  # us


def main():
    torch.manual_seed(0)
    dev = "cuda:0"
    print(f"| proj | N x K | bf16 bytes | rocBLAS us | bf16-skinny us | "
          f"fp8 us | fp8 GB/s | fp8 vs bf16-best |")
    print("|---|---|---|---|---|---|---|---|")
    tot_bf_best = tot_fp8 = 0.0
    for name, N, K in SHAPES:
        a = torch.randn(M, K, device=dev, dtype=torch.bfloat16) * 0.5
        w = torch.randn(N, K, device=dev, dtype=torch.bfloat16) * 0.02
        wf = ext().pack_weight_frag(w)
        qf, s = D.pack_weight_fp8(w)
        t_roc = timeit(lambda: torch.nn.functional.linear(a, w))
        t_bf = timeit(lambda: ext().skinny_gemm(a, wf, N, K))
        t_f8 = timeit(lambda: ext().skinny_gemm_fp8(a, qf, s, N, K))
        if os.environ.get("QSA_FP8_SWEEP") == "1":
            for waves, tiles, ntl in [(8, 1, 0), (8, 1, 1), (8, 2, 0),
                                      (8, 4, 0), (8, 4, 1), (8, 8, 0),
                                      (16, 1, 0), (16, 2, 0), (16, 4, 0),
                                      (4, 4, 0), (4, 8, 0), (2, 8, 0)]:
                if N % (16 * tiles):
                    continue
                try:
                    t = timeit(lambda: ext().skinny_gemm_fp8_probe(
                        a, qf, s, N, K, waves, tiles, ntl), reps=20)
                    print(f"    probe {name} w{waves} t{tiles} nt{ntl}: "
                          f"{t:.1f} us ({(N*K)/(t*1e-6)/1e9:.0f} GB/s)",
                          file=sys.stderr)
                except Exception as e:
                    print(f"    probe {name} w{waves} t{tiles}: {e}",
                          file=sys.stderr)
        # correctness spot check
        ref = a.float() @ D.unpack_weight_fp8(qf, s, N, K).T
        out = ext().skinny_gemm_fp8(a, qf, s, N, K).float()
        rel = (out - ref).abs().max().item() / (ref.abs().max().item() + 1e-9)
        assert rel < 2e-2, f"{name}: rel {rel}"
        bytes_bf16 = 2 * N * K
        gbs = (N * K) / (t_f8 * 1e-6) / 1e9
        best_bf = min(t_roc, t_bf)
        tot_bf_best += best_bf
        tot_fp8 += t_f8
        print(f"| {name} | {N}x{K} | {bytes_bf16/1e6:.0f} MB | "
              f"{t_roc:.1f} | {t_bf:.1f} | {t_f8:.1f} | {gbs:.0f} | "
              f"{best_bf/t_f8:.2f}x |")
    print(f"\nprojection total per decode step (x32 layers + lm_head): "
          f"bf16-best {tot_bf_best:.0f} us vs fp8 {tot_fp8:.0f} us "
          f"-> {tot_bf_best/tot_fp8:.2f}x")


if __name__ == "__main__":
    main()
