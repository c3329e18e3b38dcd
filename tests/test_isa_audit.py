"""Static ISA regression guard (tools/isa_check.py): the matrix kernels
must keep issuing MFMA instructions and no hot kernel may regress into
register spills or scratch (CDNA4 guide rule: runtime-indexed register
arrays spill silently)."""

import os
import sys

import pytest

sys.path.insert(0, os.path.join(
    os.path.dirname(os.path.dirname(os.path.abspath(__file__))), "tools"))


@pytest.fixture(scope="module")
def stats():
    import isa_check
    so = isa_check.find_so()
    if so is None or not os.path.exists(isa_check.BUNDLER):
        pytest.skip("built extension or ROCm LLVM tools unavailable")
    import tempfile
    with tempfile.TemporaryDirectory() as wd:
        out = isa_check.kernel_stats(isa_check.extract_hsacos(so, wd))
    if not out:
        pytest.skip("no gfx950 code objects extracted")
    return out


def _named(stats, frag):
    hits = {k: v for k, v in stats.items() if frag in k}
    assert hits, f"no kernel matching {frag!r}"
    return hits


def test_matrix_kernels_use_mfma(stats):
    for frag in ("paged_attn_mfma", "paged_attn_prefill", "skinny_gemm"):
        for name, st in _named(stats, frag).items():
            assert st["mfma"] >= 8, f"{name}: only {st['mfma']} v_mfma"


def test_no_spills_or_scratch_on_hot_kernels(stats):
    hot = ("paged_attn", "skinny_gemm", "rmsnorm", "rope", "swiglu",
           "softmax", "kv_append", "kv_scatter", "topk")
    for frag in hot:
        for name, st in _named(stats, frag).items():
            assert st.get("vgpr_spill_count", 0) == 0, name
            assert st.get("sgpr_spill_count", 0) == 0, name
            assert st.get("private_segment_fixed_size", 0) == 0, \
                f"{name}: scratch allocated"


def test_round2_kernels_present_and_clean(stats):
    """Round-2 kernels: fp8 GEMMs issue MFMA + the fp8 dequant path
    (v_cvt_pk_f32_fp8); the hash-join kernels exist; none spill."""
    for frag in ("skinny_gemm_fp8", "gemm_fp8_batch"):
        hits = _named(stats, frag)
        for name, st in hits.items():
            assert st["mfma"] >= 8, f"{name}: only {st['mfma']} v_mfma"
            assert st.get("vgpr_spill_count", 0) == 0, name
            assert st.get("private_segment_fixed_size", 0) == 0, name
    for frag in ("hash_build", "hash_probe", "splitk_reduce"):
        for name, st in _named(stats, frag).items():
            assert st.get("vgpr_spill_count", 0) == 0, name


def test_fp8_kernels_use_hw_dequant(stats):
    """The fp8 GEMMs must dequantize with the hardware cvt_pk_f32_fp8
    path, not a software bit-twiddling fallback."""
    import subprocess
    import tempfile

    import isa_check
    so = isa_check.find_so()
    with tempfile.TemporaryDirectory() as wd:
        cos = isa_check.extract_hsacos(so, wd)
        found = False
        for co in cos:
            dis = subprocess.run(
                [os.path.join(os.path.dirname(isa_check.BUNDLER),
                              "llvm-objdump"), "-d", co],
                capture_output=True, text=True).stdout
            if "cvt_pk_f32_fp8" in dis:
                found = True
                break
        assert found, "no v_cvt_pk_f32_fp8 in any gfx950 code object"
