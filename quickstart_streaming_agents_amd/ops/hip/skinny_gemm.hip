// Skinny-batch decode GEMM for CDNA4 (gfx950): C[M,N] = A[M,K] @ W[N,K]^T,
// M <= 32 (the decode batch), bf16 in / bf16 or f32-accum out.
//
// Decode projections are HBM-bandwidth-bound on the WEIGHT stream (the
// activations are KB-sized): speed-of-light is W bytes / 6.3 TB/s.  rocBLAS
// general-GEMM picks tiles sized for big M and lands 2-7x off that floor at
// M~24 (measured: lm_head 902us vs 167us floor).  This kernel is designed
// around the weight stream instead:
//
//   * W is PRE-PACKED once at model init into MFMA-fragment-major layout
//     [N/16, K/32, 16, 32] so one wave instruction reads one full 1 KiB
//     16x32 fragment block, perfectly coalesced, nontemporal (each byte is
//     read exactly once per step -> don't pollute L2).
//   * A (M x K, <=1 MB) stages through LDS in 32x256 chunks, cooperatively
//     loaded once per workgroup, fragments read with ds_read_b128 from
//     16B-padded rows.
//   * mfma_f32_16x16x32_bf16, 2 M-tiles (M padded to 32) x 16 N-cols per
//     wave, 4 waves = 64 N-cols per workgroup.
//   * Grid = (N/64) x SPLITK: split-K (f32 atomicAdd epilogue) keeps >=512
//     workgroups in flight for small N (wo: N=4096 -> 64x8).
//
// Numerics: f32 MFMA accumulation over the full K, identical reduction
// order to the fused-GEMM reference within a tile; split-K partials add in
// f32.
#include "common.h"

using bf16x8 = __attribute__((ext_vector_type(8))) short;
using f32x4 = __attribute__((ext_vector_type(4))) float;

#define QSA_KCH 256
#define QSA_APAD 8  // elements (16 B) of per-row LDS padding

__global__ void __launch_bounds__(256)
qsa_skinny_gemm(const unsigned short* __restrict__ A,   // [M, K] row stride lda
                const unsigned short* __restrict__ Wf,  // [N/16, K/32, 16, 32]
                unsigned short* __restrict__ Cbf,       // [M, N] (splitk==1)
                float* __restrict__ Cf32,               // [M, N] (splitk>1)
                int M, int N, long long K, long long lda, int splitk) {
  const int nblk = blockIdx.x;
  const int ks = blockIdx.y;
  const int wave = threadIdx.x >> 6;
  const int lane = threadIdx.x & 63;
  const int tid = threadIdx.x;
  const long long kchunks = K / QSA_KCH;
  const long long c0 = ks * kchunks / splitk;
  const long long c1 = (ks + 1) * kchunks / splitk;
  const int n0w = nblk * 64 + wave * 16;  // this wave's 16 output cols

  __shared__ unsigned short As[32][QSA_KCH + QSA_APAD];

  f32x4 acc0 = {0.f, 0.f, 0.f, 0.f};
  f32x4 acc1 = {0.f, 0.f, 0.f, 0.f};

  // A-fragment LDS byte address for this lane (per m-tile, per k-step):
  //   row = mt*16 + (lane&15), k = s*32 + (lane>>4)*8
  const int arow = lane & 15;
  const int akoff = (lane >> 4) * 8;

  // W stream: fragment block (n-tile, k-step) lives at
  //   ((n0w/16) * (K/32) + kk) * 512 elements; lane reads its 16 B at
  //   (lane&15)*32 + (lane>>4)*8 within the block.
  const unsigned short* wbase =
      Wf + ((long long)(n0w >> 4) * (K >> 5)) * 512 +
      (long long)((lane & 15) * 32 + (lane >> 4) * 8);

  for (long long c = c0; c < c1; ++c) {
    // ---- stage A chunk [32, 256] cooperatively (4 x 16B per thread) ----
    {
      const int piece0 = tid;  // 1024 pieces of 16 B
#pragma unroll
      for (int p = 0; p < 4; ++p) {
        const int piece = piece0 + p * 256;
        const int row = piece >> 5;           // /32
        const int off16 = piece & 31;
        uint4 v = make_uint4(0, 0, 0, 0);
        if (row < M)
          v = *reinterpret_cast<const uint4*>(
              A + (long long)row * lda + c * QSA_KCH + off16 * 8);
        *reinterpret_cast<uint4*>(&As[row][off16 * 8]) = v;
      }
    }
    __syncthreads();
    // ---- 8 k-steps of MFMA over the chunk ----
#pragma unroll
    for (int s = 0; s < 8; ++s) {
      const long long kk = c * 8 + s;
      const bf16x8 w = *reinterpret_cast<const bf16x8*>(
          __builtin_assume_aligned(wbase + kk * 512, 16));
      const bf16x8 a0 = *reinterpret_cast<const bf16x8*>(
          &As[arow][s * 32 + akoff]);
      const bf16x8 a1 = *reinterpret_cast<const bf16x8*>(
          &As[16 + arow][s * 32 + akoff]);
      acc0 = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a0, w, acc0, 0, 0, 0);
      acc1 = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a1, w, acc1, 0, 0, 0);
    }
    __syncthreads();
  }

  // ---- epilogue: C[m][n], m = mt*16 + (lane>>4)*4 + r, n = n0w + (lane&15)
  const int ncol = n0w + (lane & 15);
  const int mrow = (lane >> 4) * 4;
#pragma unroll
  for (int r = 0; r < 4; ++r) {
    const int m0 = mrow + r;
    if (splitk == 1) {
      if (m0 < M)
        Cbf[(long long)m0 * N + ncol] = f32_to_bf16(acc0[r]);
      if (16 + m0 < M)
        Cbf[(long long)(16 + m0) * N + ncol] = f32_to_bf16(acc1[r]);
    } else {
      if (m0 < M)
        atomicAdd(&Cf32[(long long)m0 * N + ncol], acc0[r]);
      if (16 + m0 < M)
        atomicAdd(&Cf32[(long long)(16 + m0) * N + ncol], acc1[r]);
    }
  }
}

// f32 -> bf16 conversion epilogue for the split-K path.
__global__ void __launch_bounds__(256)
qsa_f32_to_bf16(const float* __restrict__ src, unsigned short* __restrict__ dst,
                long long n) {
  const long long i0 = ((long long)blockIdx.x * 256 + threadIdx.x) * 8;
  if (i0 + 8 <= n) {
    float4 a = *reinterpret_cast<const float4*>(src + i0);
    float4 b = *reinterpret_cast<const float4*>(src + i0 + 4);
    uint4 o;
    o.x = f32x2_to_bf16x2(a.x, a.y);
    o.y = f32x2_to_bf16x2(a.z, a.w);
    o.z = f32x2_to_bf16x2(b.x, b.y);
    o.w = f32x2_to_bf16x2(b.z, b.w);
    *reinterpret_cast<uint4*>(dst + i0) = o;
  } else {
    for (long long i = i0; i < n; ++i) dst[i] = f32_to_bf16(src[i]);
  }
}

extern "C" void qsa_skinny_gemm_launch(const unsigned short* A,
                                       const unsigned short* Wf,
                                       unsigned short* Cbf, float* Cf32,
                                       int M, int N, long long K,
                                       long long lda, int splitk,
                                       hipStream_t stream) {
  dim3 grid(N / 64, splitk);
  hipLaunchKernelGGL(qsa_skinny_gemm, grid, dim3(256), 0, stream, A, Wf, Cbf,
                     Cf32, M, N, K, lda, splitk);
}

extern "C" void qsa_f32_to_bf16_launch(const float* src, unsigned short* dst,
                                       long long n, hipStream_t stream) {
  const long long blocks = (n / 8 + 255) / 256 + 1;
  hipLaunchKernelGGL(qsa_f32_to_bf16, dim3((unsigned)blocks), dim3(256), 0,
                     stream, src, dst, n);
}
