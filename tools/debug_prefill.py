#!/usr/bin/env python3
"""Localize flash-prefill kernel mismatches against a torch reference."""
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch

from quickstart_streaming_agents_amd.ops import ext

torch.manual_seed(0)
dev = "cuda:0"
QH, KVH, D = 4, 2, 64
import sys as _s
n = int(_s.argv[1]) if len(_s.argv) > 1 else 75
start = int(_s.argv[2]) if len(_s.argv) > 2 else 0
T = n
npages = (start + n + 63) // 64
P = npages + 2
kc = torch.randn(P, KVH, D // 8, 64, 8, device=dev, dtype=torch.bfloat16)
vc = torch.randn(P, KVH, D, 64, device=dev, dtype=torch.bfloat16)
q = torch.randn(T, QH, D, device=dev, dtype=torch.bfloat16)
bt = torch.arange(npages, dtype=torch.int32, device=dev).reshape(1, npages)
qb = list(range(0, n, 16))
qb_item = torch.zeros(len(qb), dtype=torch.int32, device=dev)
qb_pos0 = torch.tensor(qb, dtype=torch.int32, device=dev)
off_t = torch.zeros(1, dtype=torch.int32, device=dev)
start_t = torch.tensor([start], dtype=torch.int32, device=dev)
len_t = torch.tensor([n], dtype=torch.int32, device=dev)
scale = D ** -0.5
out = ext().paged_attn_prefill(q, kc, vc, bt, qb_item, qb_pos0, off_t,
                               start_t, len_t, scale)

# torch reference over the same cache
kn = kc.permute(0, 1, 3, 2, 4).reshape(P, KVH, 64, D).float()
vn = vc.permute(0, 1, 3, 2).float()
K = kn[:npages].permute(1, 0, 2, 3).reshape(KVH, npages * 64, D)
V = vn[:npages].permute(1, 0, 2, 3).reshape(KVH, npages * 64, D)
R = QH // KVH
ref = torch.zeros(T, QH, D)
for h in range(QH):
    kvh = h // R
    sc = (q[:, h].float() @ K[kvh].T) * scale          # [n, ctx]
    mask = torch.arange(npages * 64, device=dev)[None, :] > \
        (start + torch.arange(n, device=dev))[:, None]
    sc.masked_fill_(mask, float("-inf"))
    p = torch.softmax(sc, dim=-1)
    ref[:, h] = (p @ V[kvh]).cpu()
got = out.reshape(T, QH, D).float().cpu()
err = (got - ref).abs()
print("max err", err.max().item())
per_row = err.amax(dim=(1, 2))
bad = (per_row > 0.05).nonzero().flatten().tolist()
print("bad rows:", bad[:40])
per_head = err.amax(dim=(0, 2))
print("per-head max:", per_head.tolist())
if bad:
    r = bad[0]
    print("row", r, "per-head err:", err[r].amax(dim=1).tolist())
    h = int(err[r].amax(dim=1).argmax())
    grid = (err[r, h] > 0.05).int().reshape(D // 16, 16)
    print(f"head {h} bad-dim grid [dt, col]:")
    for dt in range(D // 16):
        print("  dt", dt, "".join(str(int(x)) for x in grid[dt]))
