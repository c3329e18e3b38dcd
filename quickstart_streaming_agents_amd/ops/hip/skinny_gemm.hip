// Skinny-batch decode GEMM for CDNA4 (gfx950): C[M,N] = A[M,K] @ W[N,K]^T,
// M <= 32 (the decode batch), bf16 in / bf16 out.
//
// Decode projections are HBM-bandwidth-bound on the WEIGHT stream (the
// activations are KB-sized): speed-of-light is W bytes / 6.3 TB/s.  rocBLAS
// general-GEMM tiles for big M and lands 2-7x off that floor at M~24
// (measured: lm_head 902us vs 167us BW floor).  This kernel is designed
// around the weight stream:
//
//   * W is PRE-PACKED once at model init into MFMA-fragment-major layout
//     [N/16, K/32, 16, 32]: one wave instruction reads one full 1 KiB
//     16x32 fragment block, perfectly coalesced, nontemporal (each byte
//     read exactly once per step -> don't pollute L2 with it).
//   * One workgroup per 16-column W tile, 4 waves SPLIT K inside the
//     workgroup (wave w takes k-chunks w, w+4, ...), partial accumulators
//     meet in LDS, one wave does the f32 sum + bf16 store.  Grid = N/16
//     (wo: 256 WGs, qkv: 384, wgu: 1792, lm_head: 8016) so every shape
//     fills the 256-CU chip with NO global split-K, no atomics, and no
//     epilogue kernels (the v1 design's torch.zeros + atomicAdd + convert
//     cost ~1 ms/step in launch overhead alone).
//   * A (M x K, <= 1 MB, L2/L3-resident: every WG re-reads it) loads
//     fragment-shaped straight to VGPRs; mfma_f32_16x16x32_bf16 with
//     M padded to 32 (2 m-tiles).
//
// Numerics: f32 MFMA accumulation; the 4 waves' K-partials add in f32.
#include "common.h"

using bf16x8 = __attribute__((ext_vector_type(8))) short;
using f32x4 = __attribute__((ext_vector_type(4))) float;

#define QSA_KCH 256  // k-chunk per wave-iteration (8 MFMA k-steps)

__global__ void __launch_bounds__(256)
qsa_skinny_gemm(const unsigned short* __restrict__ A,   // [M, K] row stride lda
                const unsigned short* __restrict__ Wf,  // [N/16, K/32, 16, 32]
                unsigned short* __restrict__ Cbf,       // [M, N]
                int M, int N, long long K, long long lda) {
  const int nt = blockIdx.x;              // 16-col n-tile
  const int wave = threadIdx.x >> 6;
  const int lane = threadIdx.x & 63;
  const long long kchunks = K / QSA_KCH;  // K % 256 == 0

  f32x4 acc0 = {0.f, 0.f, 0.f, 0.f};
  f32x4 acc1 = {0.f, 0.f, 0.f, 0.f};

  // A stages through LDS per SUPER-chunk of 1024 k (the 4 waves' chunks):
  // cooperative coalesced fill once, low-latency ds_read fragments after —
  // dependent per-k-step global A loads were the v2 kernel's stall.
  // Row pad 8 elements (16 B) spreads the 16-row fragment read groups
  // across banks.
  __shared__ unsigned short As[32][1024 + 8];

  // A fragment LDS address pieces: row mt*16 + (lane&15),
  // k-offset = wave*256 + s*32 + (lane>>4)*8 within the super-chunk.
  const int arow = lane & 15;
  const int akoff = (lane >> 4) * 8;

  // W stream: block (nt, kk) at ((nt*(K/32) + kk) * 512) elements; this
  // lane's 16 B at (lane&15)*32 + (lane>>4)*8 inside the block.
  const unsigned short* wbase =
      Wf + (long long)nt * (K >> 5) * 512 +
      (long long)((lane & 15) * 32 + (lane >> 4) * 8);

  const long long nsuper = kchunks / 4;   // K % 1024 == 0 (K % 256 == 0 and
  const long long ktail = nsuper * 4;     // tail chunks handled separately)
  for (long long sc = 0; sc < nsuper; ++sc) {
    // ---- cooperative stage: A[0:32][sc*1024 : +1024] (64 KiB) ----------
    {
      const int tid = (int)threadIdx.x;
      // 2048 pieces of 16 B; thread t fills pieces t, t+256, ...
#pragma unroll
      for (int p = 0; p < 8; ++p) {
        const int piece = tid + p * 256;
        const int row = piece >> 6;            // 64 pieces per row
        const int off16 = piece & 63;
        uint4 v = make_uint4(0, 0, 0, 0);
        if (row < M)
          v = *reinterpret_cast<const uint4*>(
              A + (long long)row * lda + sc * 1024 + off16 * 8);
        *reinterpret_cast<uint4*>(&As[row][off16 * 8]) = v;
      }
    }
    __syncthreads();
    const long long k0 = sc * 1024 + wave * QSA_KCH;
#pragma unroll
    for (int s = 0; s < 8; ++s) {
      const long long kk = (k0 >> 5) + s;
      const bf16x8 w = *reinterpret_cast<const bf16x8*>(
          __builtin_assume_aligned(wbase + kk * 512, 16));
      const int la = wave * QSA_KCH + s * 32 + akoff;
      const bf16x8 a0 = *reinterpret_cast<const bf16x8*>(&As[arow][la]);
      const bf16x8 a1 = *reinterpret_cast<const bf16x8*>(&As[16 + arow][la]);
      acc0 = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a0, w, acc0, 0, 0, 0);
      acc1 = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a1, w, acc1, 0, 0, 0);
    }
    __syncthreads();
  }
  // ---- K tail (kchunks % 4 != 0): direct global fragments --------------
  if (ktail < kchunks) {
    const bf16x8 zero8 = {0, 0, 0, 0, 0, 0, 0, 0};
    const bool row0_ok = arow < M;
    const bool row1_ok = 16 + arow < M;
    const unsigned short* a0base = A + (long long)arow * lda + akoff;
    const unsigned short* a1base = A + (long long)(16 + arow) * lda + akoff;
    for (long long c = ktail + wave; c < kchunks; c += 4) {
      const long long k0 = c * QSA_KCH;
#pragma unroll
      for (int s = 0; s < 8; ++s) {
        const long long kk = (k0 >> 5) + s;
        const bf16x8 w = *reinterpret_cast<const bf16x8*>(
            __builtin_assume_aligned(wbase + kk * 512, 16));
        const long long ak = k0 + s * 32;
        const bf16x8 a0 = row0_ok
            ? *reinterpret_cast<const bf16x8*>(a0base + ak) : zero8;
        const bf16x8 a1 = row1_ok
            ? *reinterpret_cast<const bf16x8*>(a1base + ak) : zero8;
        acc0 = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a0, w, acc0, 0, 0, 0);
        acc1 = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a1, w, acc1, 0, 0, 0);
      }
    }
  }

  // ---- cross-wave K-reduction in LDS ----------------------------------
  __shared__ float red[4][64][8];  // wave, lane, 8 acc f32
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
      const float v0 = red[0][lane][r] + red[1][lane][r] +
                       red[2][lane][r] + red[3][lane][r];
      const float v1 = red[0][lane][4 + r] + red[1][lane][4 + r] +
                       red[2][lane][4 + r] + red[3][lane][4 + r];
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
  hipLaunchKernelGGL(qsa_skinny_gemm, dim3(N / 16), dim3(256), 0, stream, A,
                     Wf, Cbf, M, N, K, lda);
}
