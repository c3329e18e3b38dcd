import os, sys, time
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch

t = torch.cuda.tunable
t.enable(True)
t.tuning_enable(True)
# keep per-shape tuning bounded
t.set_max_tuning_duration(100)      # ms per candidate set
t.set_max_tuning_iterations(30)

dev = "cuda:0"
SHAPES = [(6144, 4096), (4096, 4096), (28672, 4096), (4096, 14336)]
BUCKETS = [4096, 8192, 12288, 16384, 24576, 40960, 49152, 57344]
t0 = time.perf_counter()
for M in BUCKETS:
    for N, K in SHAPES:
        x = torch.randn(M, K, device=dev, dtype=torch.bfloat16)
        w = torch.randn(N, K, device=dev, dtype=torch.bfloat16)
        for _ in range(3):
            torch.nn.functional.linear(x, w)
        torch.cuda.synchronize()
        print(f"tuned {N}x{M}x{K}  ({time.perf_counter()-t0:.0f}s)", flush=True)
t.write_file("gpurun_out/tuned_new.csv")
print("wrote gpurun_out/tuned_new.csv")
