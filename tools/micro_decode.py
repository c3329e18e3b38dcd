#!/usr/bin/env python3
"""Decode-step microbenchmark: N graph-replayed decode steps of the
flagship model at fixed batch/context, no prefill in the timed region.

Usage: python tools/micro_decode.py [--model llama3-8b] [--batch 24]
         [--ctx 512] [--steps 200]
Prints ms/step, tokens/s, and the HBM-bound floor (weights+KV bytes / 8 TB/s)
for calibration.  Run under rocprofv3 for per-kernel attribution.
"""

import argparse
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch  # noqa: E402


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--model", default="llama3-8b")
    p.add_argument("--batch", type=int, default=24)
    p.add_argument("--ctx", type=int, default=512)
    p.add_argument("--steps", type=int, default=200)
    p.add_argument("--warmup", type=int, default=20)
    p.add_argument("--eager", action="store_true")
    args = p.parse_args()

    from quickstart_streaming_agents_amd.models import build_model
    from quickstart_streaming_agents_amd.models.serve import Engine

    assert torch.cuda.is_available()
    model = build_model(args.model, device="cuda:0")
    total = args.ctx + args.steps + args.warmup + 8
    eng = Engine(model, max_batch=args.batch, max_seq_len=total)
    if args.eager:
        eng.use_graph = False
    torch.manual_seed(0)
    prompts = [torch.randint(16, 2000, (args.ctx,)).tolist()
               for _ in range(args.batch)]
    seqs = [eng.submit(pr, args.warmup + args.steps) for pr in prompts]
    # prefill + warmup decode
    while eng.pending:
        n = len(eng.pending)
        eng._admit()
        if len(eng.pending) == n:
            raise MemoryError("cannot admit")
    batch = [s for s in eng.running]
    if eng.use_graph:
        eng._decode_run_graph(batch, args.warmup)
    else:
        for _ in range(args.warmup):
            eng.step()
        batch = [s for s in eng.running]
    torch.cuda.synchronize()

    t0 = time.perf_counter()
    if eng.use_graph:
        eng._decode_run_graph(batch, args.steps)
    else:
        for _ in range(args.steps):
            eng.step()
    torch.cuda.synchronize()
    dt = time.perf_counter() - t0

    c = model.cfg
    w_bytes = 2 * (c.vocab_size * c.hidden * 2 +
                   c.n_layers * (model.qkv_dim * c.hidden +
                                 c.hidden * model.n_q * c.d_head +
                                 3 * model.ffn_local * c.hidden))
    kv_bytes = 2 * 2 * c.n_layers * model.n_kv * c.d_head * \
        args.batch * (args.ctx + args.steps / 2)
    floor_ms = (w_bytes + kv_bytes) / 8e12 * 1e3
    ms = dt / args.steps * 1e3
    print(f"model={args.model} batch={args.batch} ctx={args.ctx} "
          f"graph={eng.use_graph}")
    print(f"ms_per_step={ms:.3f} tokens_per_s={args.batch / ms * 1e3:.0f} "
          f"hbm_floor_ms={floor_ms:.3f} frac_of_sol={floor_ms / ms:.2%}")


if __name__ == "__main__":
    main()
