import os, sys; sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import time
import torch
from quickstart_streaming_agents_amd.ops import dispatch as D, ext

def timeit(fn, reps=30):
    for _ in range(5): fn()
    torch.cuda.synchronize(); t0 = time.perf_counter()
    for _ in range(reps): fn()
    torch.cuda.synchronize(); return (time.perf_counter()-t0)/reps*1e6

torch.manual_seed(0)
dev = "cuda:0"
for M in (192, 128, 64):
    print(f"== M={M}")
    for name, N, K in [("qkv",6144,4096),("wo",4096,4096),("wgu",28672,4096),("wdown",4096,14336)]:
        a = torch.randn(M, K, device=dev, dtype=torch.bfloat16)*0.5
        w = torch.randn(N, K, device=dev, dtype=torch.bfloat16)*0.02
        qf, s = D.pack_weight_fp8(w)
        t_roc = timeit(lambda: torch.nn.functional.linear(a, w))
        res = []
        for sk in (1,2,4):
            if K % (64*sk): continue
            t = timeit(lambda: ext().gemm_fp8_batch(a, qf, s, N, K, sk))
            res.append((sk, t))
        best = min(res, key=lambda x: x[1])
        floor = N*K/6.3e12*1e6
        print(f"  {name:6s} roc={t_roc:7.1f}us  fp8batch={'/'.join(f'sk{k}:{t:.1f}' for k,t in res)}  best={best[1]:.1f} (sk{best[0]})  fp8floor={floor:.1f}us")
