import os, sys, time
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch

t = torch.cuda.tunable
t.enable(True)
t.tuning_enable(True)
t.set_max_tuning_duration(50)
t.set_max_tuning_iterations(10)

dev = "cuda:0"
SHAPES = [(6144, 4096), (4096, 4096), (28672, 4096), (4096, 14336)]
BUCKETS = [16384, 24576, 40960, 49152]
OUT = "gpurun_out/tuned_new.csv"


def dump():
    vals = t.get_validators()
    res = t.get_results()
    with open(OUT, "w") as fh:
        for k, v in vals:
            fh.write(f"Validator,{k},{v}\n")
        for op, params, sol, ms in res:
            fh.write(f"{op},{params},{sol},{ms}\n")


t0 = time.perf_counter()
for M in BUCKETS:
    for N, K in SHAPES:
        x = torch.randn(M, K, device=dev, dtype=torch.bfloat16)
        w = torch.randn(N, K, device=dev, dtype=torch.bfloat16)
        for _ in range(3):
            torch.nn.functional.linear(x, w)
        torch.cuda.synchronize()
        dump()   # persist incrementally (timeout-safe)
        print(f"tuned {N}x{M}x{K}  ({time.perf_counter()-t0:.0f}s)",
              flush=True)
print("done")
