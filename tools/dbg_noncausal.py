import os, sys; sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
from quickstart_streaming_agents_amd.ops import ext, dispatch as D

torch.manual_seed(0)
dev = "cuda:0"
QH = KVH = 8
Dh = 64
lens = [37, 64, 5, 130]       # ragged items
n = len(lens)
L = max(lens)
ppi = (L + 63) // 64
scale = Dh ** -0.5

T = n * L
q = torch.randn(T, QH, Dh, device=dev, dtype=torch.bfloat16) * 0.5
k = torch.randn(T, KVH, Dh, device=dev, dtype=torch.bfloat16) * 0.5
v = torch.randn(T, KVH, Dh, device=dev, dtype=torch.bfloat16) * 0.5

kc = torch.zeros(n * ppi, KVH, Dh // 8, 64, 8, device=dev, dtype=torch.bfloat16)
vc = torch.zeros(n * ppi, KVH, Dh, 64, device=dev, dtype=torch.bfloat16)
slots = torch.arange(n * ppi * 64, dtype=torch.int32, device=dev) \
    .reshape(n, ppi * 64)[:, :L].reshape(-1).contiguous()
D.kv_scatter(k, v, kc, vc, slots)
bt = torch.arange(n * ppi, dtype=torch.int32, device=dev).reshape(n, ppi).contiguous()
qb_item, qb_pos0 = [], []
for i in range(n):
    for p0 in range(0, lens[i], 16):
        qb_item.append(i); qb_pos0.append(p0)
mk = lambda x: torch.tensor(x, dtype=torch.int32, device=dev)
for causal in (True, False):
    out = ext().paged_attn_prefill(
        q, kc, vc, bt, mk(qb_item), mk(qb_pos0),
        mk([i * L for i in range(n)]), mk([0] * n), mk(lens), scale, causal)
    # reference
    bad = 0
    for i in range(n):
        li = lens[i]
        qi = q[i*L:i*L+li].float()   # [li, H, D]
        ki = k[i*L:i*L+li].float()
        vi = v[i*L:i*L+li].float()
        sc = torch.einsum("qhd,khd->hqk", qi, ki) * scale
        if causal:
            mask = torch.triu(torch.ones(li, li, device=dev, dtype=torch.bool), 1)
            sc = sc.masked_fill(mask.unsqueeze(0), float("-inf"))
        p = sc.softmax(-1)
        ref = torch.einsum("hqk,khd->qhd", p, vi).reshape(li, QH * Dh)
        got = out[i*L:i*L+li].float()
        err = (got - ref).abs().max().item()
        rel = err / (ref.abs().max().item() + 1e-9)
        if rel > 2e-2 or got.isnan().any():
            bad += 1
            print(f"causal={causal} item {i} len {li}: rel {rel:.4f} nan={got.isnan().any().item()}")
            # locate first bad row
            rower = (got - ref).abs().amax(1)
            bi = int(rower.argmax())
            print("   worst row", bi, "err", rower[bi].item(), "got", got[bi,:4].tolist(), "ref", ref[bi,:4].tolist())
    print(f"causal={causal}: {'OK' if bad == 0 else f'{bad} BAD items'}")
