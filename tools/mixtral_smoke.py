#!/usr/bin/env python3
"""Config-5 smoke: full-size Mixtral-8x7B decode on one MI355X."""
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch  # noqa: E402

from quickstart_streaming_agents_amd.models import build_model  # noqa: E402
from quickstart_streaming_agents_amd.models.serve import Engine  # noqa: E402

t0 = time.perf_counter()
model = build_model("mixtral-8x7b", device="cuda:0")
torch.cuda.synchronize()
print(f"init {time.perf_counter()-t0:.1f}s; "
      f"mem {torch.cuda.memory_allocated()/2**30:.1f} GiB")
eng = Engine(model, max_batch=32, max_seq_len=1024)
prompts = [list(range(5, 260)) for _ in range(32)]
t0 = time.perf_counter()
outs = eng.generate_batch(prompts, [32] * 32)
torch.cuda.synchronize()
dt = time.perf_counter() - t0
st = eng.stats
print(f"32 seqs x 32 tokens in {dt:.2f}s; decode_steps={st.decode_steps} "
      f"decode_tokens={st.decode_tokens} "
      f"tokens/s={st.decode_tokens/dt:.0f}")
assert all(len(o) == 32 for o in outs)
print("mixtral smoke ok")
