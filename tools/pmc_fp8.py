import os, sys; sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
from quickstart_streaming_agents_amd.ops import dispatch as D, ext
torch.manual_seed(0)
dev = "cuda:0"
# fp8 skinny at the llama shapes (M=24 decode), a few reps each
for name, N, K in [("qkv",6144,4096),("wgu",28672,4096),("lm_head",128256,4096)]:
    a = torch.randn(24, K, device=dev, dtype=torch.bfloat16)*0.5
    w = torch.randn(N, K, device=dev, dtype=torch.bfloat16)*0.02
    qf, s = D.pack_weight_fp8(w)
    for _ in range(10):
        ext().skinny_gemm_fp8(a, qf, s, N, K)
torch.cuda.synchronize()
print("done")
