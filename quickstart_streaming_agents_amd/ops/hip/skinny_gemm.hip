// Skinny-batch decode GEMM for CDNA4 (gfx950): C[M,N] = A[M,K] @ W[N,K]^T,
// M <= 32 (the decode batch), bf16 in / bf16 out.
// Serves the K4 agent-LLM decode projections.
//
// Decode projections are HBM-bandwidth-bound on the WEIGHT stream (the
// activations are KB-sized): speed-of-light is W bytes / 6.3 TB/s.  rocBLAS
// general-GEMM tiles for big M and loses on the latency-critical small
// shapes at M~24.  Design:
//
//   * W is PRE-PACKED once at model init into MFMA-fragment-major layout
//     [N/16, K/32, 16, 32]: one wave instruction reads one full 1 KiB
//     16x32 fragment block, perfectly coalesced, nontemporal (each byte
//     read exactly once per step -> don't displace L2 lines; guide row
//     "nt-weights").
//   * One workgroup per 16-column W tile, WAVES waves SPLIT K inside the
//     workgroup, partial accumulators meet in LDS, wave 0 reduces + does
//     the bf16 store.  Grid = N/16 (wo: 256 WGs, qkv: 384, wgu: 1792,
//     lm_head: 8016): every decode shape fills the 256-CU chip with no
//     global split-K, no atomics, no epilogue kernels.
//   * A (M x K, <= 1 MB, L2-resident: every WG re-reads it) loads
//     fragment-shaped straight to VGPRs (guide: "GEMV / M<=16 decode
//     weights: load straight to VGPRs, deep unroll" — LDS staging is pure
//     overhead at this operand size); mfma_f32_16x16x32_bf16, M padded to
//     32 (2 m-tiles).
//
// The VARIANT template parameter exists for ablation probes from
// tools/kernel_bench.py (1 = W stream + MFMA only, 2 = A loads + MFMA
// only); production always runs VARIANT 0.
#include "common.h"

using bf16x8 = __attribute__((ext_vector_type(8))) short;
using f32x4 = __attribute__((ext_vector_type(4))) float;

#define QSA_KCH 256  // k-chunk per wave-iteration (8 MFMA k-steps)

template <int WAVES, bool NT, int VARIANT>
__global__ void __launch_bounds__(WAVES * 64)
qsa_skinny_gemm_t(const unsigned short* __restrict__ A,   // [M,K] stride lda
                  const unsigned short* __restrict__ Wf,  // [N/16,K/32,16,32]
                  unsigned short* __restrict__ Cbf,       // [M, N]
                  int M, int N, long long K, long long lda) {
  const int nt = blockIdx.x;              // 16-col n-tile
  const int wave = threadIdx.x >> 6;
  const int lane = threadIdx.x & 63;
  const long long kchunks = K / QSA_KCH;  // K % 256 == 0

  f32x4 acc0 = {0.f, 0.f, 0.f, 0.f};
  f32x4 acc1 = {0.f, 0.f, 0.f, 0.f};

  // Out-of-batch rows CLAMP to row M-1 and load garbage that is never
  // stored (per-row MFMA outputs are independent; the store is guarded).
  // A conditional `ok ? load : zero` here would be the guide's .s-level
  // trap (c): hipcc branches around each load and drains vmcnt(0) per
  // element — measured 1.5-2.5x on this kernel.
  const int arow = lane & 15;
  const int akoff = (lane >> 4) * 8;
  const int r0 = min(arow, M - 1);
  const int r1 = min(16 + arow, M - 1);
  const unsigned short* a0base = A + (long long)r0 * lda + akoff;
  const unsigned short* a1base = A + (long long)r1 * lda + akoff;

  // W stream: block (nt, kk) at ((nt*(K/32) + kk) * 512) elements; this
  // lane's 16 B at (lane&15)*32 + (lane>>4)*8 inside the block.
  const unsigned short* wbase =
      Wf + (long long)nt * (K >> 5) * 512 +
      (long long)((lane & 15) * 32 + (lane >> 4) * 8);

  const bf16x8 zero8 = {0, 0, 0, 0, 0, 0, 0, 0};
  for (long long c = wave; c < kchunks; c += WAVES) {
    const long long k0 = c * QSA_KCH;
#pragma unroll
    for (int s = 0; s < 8; ++s) {
      const long long kk = (k0 >> 5) + s;
      bf16x8 w;
      if (VARIANT == 2) {
        w = zero8;
      } else if (NT) {
        w = __builtin_nontemporal_load(
            reinterpret_cast<const bf16x8*>(wbase + kk * 512));
      } else {
        w = *reinterpret_cast<const bf16x8*>(
            __builtin_assume_aligned(wbase + kk * 512, 16));
      }
      const long long ak = k0 + s * 32;
      bf16x8 a0 = zero8, a1 = zero8;
      if (VARIANT != 1) {
        a0 = *reinterpret_cast<const bf16x8*>(a0base + ak);
        a1 = *reinterpret_cast<const bf16x8*>(a1base + ak);
      }
      acc0 = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a0, w, acc0, 0, 0, 0);
      acc1 = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a1, w, acc1, 0, 0, 0);
    }
  }

  // ---- cross-wave K-reduction in LDS ----------------------------------
  __shared__ float red[WAVES][64][8];
#pragma unroll
  for (int r = 0; r < 4; ++r) {
    red[wave][lane][r] = acc0[r];
    red[wave][lane][4 + r] = acc1[r];
  }
  __syncthreads();
  if (wave == 0) {
    const int ncol = nt * 16 + (lane & 15);
    const int mrow = (lane >> 4) * 4;
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      float v0 = 0.f, v1 = 0.f;
#pragma unroll
      for (int wv = 0; wv < WAVES; ++wv) {
        v0 += red[wv][lane][r];
        v1 += red[wv][lane][4 + r];
      }
      const int m0 = mrow + r;
      if (m0 < M) Cbf[(long long)m0 * N + ncol] = f32_to_bf16(v0);
      if (16 + m0 < M)
        Cbf[(long long)(16 + m0) * N + ncol] = f32_to_bf16(v1);
    }
  }
}

extern "C" void qsa_skinny_gemm_launch(const unsigned short* A,
                                       const unsigned short* Wf,
                                       unsigned short* Cbf, int M, int N,
                                       long long K, long long lda,
                                       hipStream_t stream) {
  // 8 waves, plain (non-nt) W loads: the measured best full-kernel config
  // (tools/kernel_bench.py sweep; nt helps only the W-isolated stream).
  hipLaunchKernelGGL((qsa_skinny_gemm_t<8, false, 0>), dim3(N / 16),
                     dim3(512), 0, stream, A, Wf, Cbf, M, N, K, lda);
}

// Ablation/tuning probe for tools/kernel_bench.py.
extern "C" void qsa_skinny_gemm_probe_launch(
    const unsigned short* A, const unsigned short* Wf, unsigned short* Cbf,
    int M, int N, long long K, long long lda, int waves, int nt, int variant,
    hipStream_t stream) {
  dim3 grid(N / 16);
#define QSA_CASE(W, NTB, V)                                            \
  if (waves == W && nt == NTB && variant == V) {                       \
    hipLaunchKernelGGL((qsa_skinny_gemm_t<W, NTB, V>), grid,           \
                       dim3(W * 64), 0, stream, A, Wf, Cbf, M, N, K,   \
                       lda);                                           \
    return;                                                            \
  }
  QSA_CASE(4, true, 0) QSA_CASE(4, false, 0) QSA_CASE(4, true, 1)
  QSA_CASE(4, true, 2) QSA_CASE(8, true, 0) QSA_CASE(8, false, 0)
  QSA_CASE(8, true, 1) QSA_CASE(8, true, 2) QSA_CASE(2, true, 0)
  QSA_CASE(1, true, 0)
#undef QSA_CASE
}
