import os, sys; sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import time
from quickstart_streaming_agents_amd.labs import datagen
from quickstart_streaming_agents_amd.models.encoder import EmbeddingEncoder

enc = EmbeddingEncoder(device="cuda:0")
docs = [d["chunk"] for d in datagen.lab2_documents(n_chunks=64)]
texts = (docs * 64)[:4096]
enc.embed_batch(texts[:256])  # warmup
import torch; torch.cuda.synchronize()
t0 = time.perf_counter()
out = enc.embed_batch(texts)
torch.cuda.synchronize()
dt = time.perf_counter() - t0
ntok = sum(len(enc.tokenizer.encode(t)[:512]) for t in texts)
c = enc.cfg
# per-token forward FLOPs (GEMM-dominated): qkv+wo (attn_dim) + ffn + proj
flops_tok = 2 * c.hidden * (4 * c.attn_dim + 3 * c.ffn) * c.n_layers \
    + 2 * c.hidden * c.out_dim
print(f"texts/s: {len(texts)/dt:.0f}  tokens/s: {ntok/dt:.0f}  "
      f"model TFLOP/s: {ntok/dt*flops_tok/1e12:.1f} "
      f"(bf16 MFMA peak 2495; small-GEMM-bound is expected)")
print(f"avg tokens/text: {ntok/len(texts):.1f}  batch wall: {dt*1e3:.0f} ms")
