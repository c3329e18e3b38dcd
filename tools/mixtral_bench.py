import os, sys, time
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
from quickstart_streaming_agents_amd.models.mixtral import MixtralConfig, MixtralModel
from quickstart_streaming_agents_amd.models.serve import Engine

t0 = time.perf_counter()
model = MixtralModel(MixtralConfig.preset("mixtral-8x7b"), device="cuda:0", seed=0)
torch.cuda.synchronize()
print(f"init {time.perf_counter()-t0:.0f}s  mem {torch.cuda.memory_allocated()/1e9:.0f} GB")
B, CTX, NEW = 24, 512, 64
eng = Engine(model, max_batch=B, max_seq_len=1024)
prompts = [[1] + [(17 * i + j) % 30000 + 16 for j in range(CTX - 1)] for i in range(B)]
eng.generate_batch([list(p) for p in prompts], [4] * B)  # warmup+capture
torch.cuda.synchronize()
t0 = time.perf_counter()
outs = eng.generate_batch([list(p) for p in prompts], [NEW] * B)
torch.cuda.synchronize()
dt = time.perf_counter() - t0
assert all(len(o) == NEW for o in outs)
print(f"mixtral-8x7b 1xMI355X: batch {B}, ctx {CTX}, {NEW} new: "
      f"{B*NEW/dt:.0f} decode tok/s incl prefill, wall {dt:.2f}s, "
      f"graph={'on' if eng._graph is not None else 'off'}")
