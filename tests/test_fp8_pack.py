"""CPU tests for the fp8 (e4m3, per-channel W8A16) weight packing used by
the decode GEMM (ops/hip/skinny_gemm_fp8.hip)."""

import torch

from quickstart_streaming_agents_amd.ops import dispatch as D


def test_pack_unpack_round_trip_layout():
    torch.manual_seed(0)
    w = (torch.randn(64, 512) * 0.02).to(torch.bfloat16)
    qf, s = D.pack_weight_fp8(w)
    assert qf.dtype == torch.uint8 and qf.numel() == w.numel()
    assert s.shape == (64,) and s.dtype == torch.float32
    wd = D.unpack_weight_fp8(qf, s, 64, 512)
    # e4m3 with per-channel scales: <= ~6.5% relative per element
    rel = ((wd - w.float()).abs() /
           w.float().abs().clamp(min=1e-6)).max().item()
    assert rel < 0.07, rel


def test_pack_layout_byte_addressing():
    """The kernel reads lane (r, kq)'s 16 bytes at block offset
    r*64 + kq*16: bytes 0..7 = k[kq*8:+8], 8..15 = k[32+kq*8:+8]."""
    n, k = 16, 256
    w = torch.zeros(n, k, dtype=torch.bfloat16)
    # distinct, exactly fp8-representable values (small ints)
    for r in range(n):
        for kk in range(k):
            w[r, kk] = float((r * 7 + kk) % 15 - 7)
    qf, s = D.pack_weight_fp8(w)
    blocks = qf.reshape(k // 64, 16, 4, 16)  # [kb, r, kq, 16B]
    wd = D.unpack_weight_fp8(qf, s, n, k)
    for kb in range(k // 64):
        for r in range(0, 16, 5):
            for kq in range(4):
                got = blocks[kb, r, kq].view(torch.float8_e4m3fn).float() \
                    * s[r]
                lo = wd[r, kb * 64 + kq * 8: kb * 64 + kq * 8 + 8]
                hi = wd[r, kb * 64 + 32 + kq * 8: kb * 64 + 32 + kq * 8 + 8]
                assert torch.equal(got[:8], lo)
                assert torch.equal(got[8:], hi)


def test_cpu_linear_fp8_close_to_bf16():
    torch.manual_seed(1)
    M, N, K = 8, 128, 256
    w = (torch.randn(N, K) * 0.02).to(torch.bfloat16)
    x = (torch.randn(M, K) * 0.5).to(torch.bfloat16)
    qf, s = D.pack_weight_fp8(w)
    out = D.skinny_linear_fp8(x, qf, s, N, K)
    ref = x.float() @ w.float().T
    rel = (out.float() - ref).abs().max().item() / \
        (ref.abs().std().item() + 1e-6)
    assert rel < 0.5, rel
