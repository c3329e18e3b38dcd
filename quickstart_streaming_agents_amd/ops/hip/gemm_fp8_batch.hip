// Batched-M fp8 weight-streaming GEMM for CDNA4 (gfx950):
// C[M,N] = A[M,K] @ (s[N] * Q[N,K])^T, 32 < M <= 256, A bf16, Q fp8.
//
// The flagship decode runs continuous batches of ~192 sequences; rocBLAS
// bf16 runs these projections ~2.5x above the weight-stream floor
// (profiles/README.md round-2 breakdown).  This kernel streams the fp8
// weights EXACTLY ONCE (half the bf16 bytes) and stages the activations
// through LDS:
//
//   * Workgroup = 4 waves, C block [16*M_FRAGS, 64]: wave w owns the
//     16-column n-tile nt = blockIdx.x*4 + w for the FULL M range, so
//     its weight fragments are private (one 16-B dwordx4 per 64-k step,
//     the fragment-pair-major layout of skinny_gemm_fp8.hip) while the
//     A tile in LDS is shared by all 4 waves.
//   * A tile [16*M_FRAGS][64] bf16, row-padded to 80 elements: the
//     (40*row + 4*hi) mod 64 bank pattern makes both the cooperative
//     ds_write_b128 fill and the per-fragment ds_read_b128 reads
//     conflict-free (checked per the LDS banking table).
//   * SPLITK > 1 (grid.y) splits the K range for the small-N shapes
//     (qkv/wo/wdown at N/64 < 256 workgroups): each split accumulates
//     into its own f32 slab; qsa_splitk_reduce sums the slabs, applies
//     the per-channel scale and casts to bf16.
//   * In-kernel dequant is the exact cvt_pk_f32_fp8 + v_perm path shared
//     with the skinny kernel (fp8 -> bf16 is lossless).
//
// At M=192 these shapes sit at the compute/bandwidth crossover
// (wgu: 45 GFLOP ~ 22 us at the bf16 MFMA rate vs 18.6 us fp8 stream),
// so the ceiling is ~max(stream, MFMA) — the kernel's job is to pin the
// stream at 1x W bytes and keep the MFMA pipe fed from LDS.
#include "common.h"

using bf16x8 = __attribute__((ext_vector_type(8))) short;
using f32x2v = __attribute__((ext_vector_type(2))) float;
using f32x4 = __attribute__((ext_vector_type(4))) float;
using u32x4 = __attribute__((ext_vector_type(4))) unsigned int;

#define QSA_GB_PAD 80   // LDS row pitch (elements) for the A tile

__device__ __forceinline__ bf16x8 qsa_gb_cvt(unsigned int a,
                                             unsigned int b) {
  f32x2v f01 = __builtin_amdgcn_cvt_pk_f32_fp8(a, false);
  f32x2v f23 = __builtin_amdgcn_cvt_pk_f32_fp8(a, true);
  f32x2v f45 = __builtin_amdgcn_cvt_pk_f32_fp8(b, false);
  f32x2v f67 = __builtin_amdgcn_cvt_pk_f32_fp8(b, true);
  union { unsigned int u; float f; } u0, u1;
  union { unsigned int u[4]; bf16x8 v; } out;
  u0.f = f01.x; u1.f = f01.y;
  out.u[0] = __builtin_amdgcn_perm(u1.u, u0.u, 0x07060302u);
  u0.f = f23.x; u1.f = f23.y;
  out.u[1] = __builtin_amdgcn_perm(u1.u, u0.u, 0x07060302u);
  u0.f = f45.x; u1.f = f45.y;
  out.u[2] = __builtin_amdgcn_perm(u1.u, u0.u, 0x07060302u);
  u0.f = f67.x; u1.f = f67.y;
  out.u[3] = __builtin_amdgcn_perm(u1.u, u0.u, 0x07060302u);
  return out.v;
}

template <int M_FRAGS, int SPLITK, int NT>
__global__ void __launch_bounds__(256)
qsa_gemm_fp8_batch(const unsigned short* __restrict__ A,  // [M,K] bf16
                   const u32x4* __restrict__ Qf,  // fp8 fragment stream
                   const float* __restrict__ scale,       // [N]
                   unsigned short* __restrict__ Cbf,      // [M,N] (SPLITK=1)
                   float* __restrict__ ws,     // [SPLITK,M,N] (SPLITK>1)
                   int M, int N, long long K, long long lda) {
  constexpr int R = 16 * M_FRAGS;          // A tile rows
  const int wave = threadIdx.x >> 6;
  const int lane = threadIdx.x & 63;
  const int tid = threadIdx.x;
  // this wave's NT 16-col n-tiles (NT=2 halves the per-WG A re-read,
  // the L2-bandwidth bound at 64 output columns per workgroup)
  const int nt16 = blockIdx.x * 4 * NT + wave * NT;
  const long long ks = (long long)blockIdx.y * (K / SPLITK);
  const long long ke = ks + K / SPLITK;

  // double-buffered A tile: iteration kb computes on buf[kb&1] while
  // the NEXT tile's global loads (issued before the compute) land in
  // registers and drain to buf[(kb+1)&1] after the MFMA sweep — one
  // barrier per k-step, memory latency hidden under the MFMA stream
  __shared__ unsigned short atile[2][R * QSA_GB_PAD];

  f32x4 acc[NT][M_FRAGS];
#pragma unroll
  for (int t = 0; t < NT; ++t)
#pragma unroll
    for (int m = 0; m < M_FRAGS; ++m)
      acc[t][m] = (f32x4){0.f, 0.f, 0.f, 0.f};

  // W stream bases for this wave's n-tiles (16 B per 64-k block; layout
  // identical to skinny_gemm_fp8.hip)
  const u32x4* qbase[NT];
#pragma unroll
  for (int t = 0; t < NT; ++t)
    qbase[t] = Qf + ((long long)(nt16 + t) * (K >> 6)) * 64 +
               (long long)((lane & 15) * 4 + (lane >> 4));

  // cooperative A fill map: chunk c covers row c/8, 16 B at col (c%8)*8
  constexpr int CHUNKS = R * 8;            // 16-B chunks per 64-k tile
  constexpr int PER_T = CHUNKS / 256;

  const int col = lane & 15;
  const int hi = lane >> 4;
  const int frow = tid >> 3;               // this thread's fill row
  const int fc8 = (tid & 7) * 8;           // and 16-B column offset
  // (PER_T chunks per thread: rows frow + i*32, same 16-B column)

  const long long kb0 = ks >> 6, kb1 = ke >> 6;
  // prologue: fill buf0 with tile kb0, and take kb0's W fragment
  {
    const long long k0 = kb0 << 6;
#pragma unroll
    for (int i = 0; i < PER_T; ++i) {
      const int row = frow + i * 32;
      const int arow = min(row, M - 1);
      const bf16x8 v = *reinterpret_cast<const bf16x8*>(
          A + (long long)arow * lda + k0 + fc8);
      *reinterpret_cast<bf16x8*>(&atile[0][row * QSA_GB_PAD + fc8]) = v;
    }
  }
  u32x4 q[NT];
#pragma unroll
  for (int t = 0; t < NT; ++t)
    q[t] = __builtin_nontemporal_load(qbase[t] + kb0 * 64);
  __syncthreads();

  for (long long kb = kb0; kb < kb1; ++kb) {
    const int cur = (int)((kb - kb0) & 1);
    // ---- issue NEXT tile's global loads (A and W, no wait) ------------
    bf16x8 stage[PER_T];
    u32x4 q_next[NT];
    const bool have_next = (kb + 1) < kb1;
    if (have_next) {
      const long long kn = (kb + 1) << 6;
#pragma unroll
      for (int t = 0; t < NT; ++t)
        q_next[t] = __builtin_nontemporal_load(qbase[t] + (kb + 1) * 64);
#pragma unroll
      for (int i = 0; i < PER_T; ++i) {
        const int arow = min(frow + i * 32, M - 1);
        stage[i] = *reinterpret_cast<const bf16x8*>(
            A + (long long)arow * lda + kn + fc8);
      }
    }
    // ---- this wave's W fragments + MFMA sweep over the m-frags --------
    // 2-deep software pipeline on the LDS reads: fragment m's reads are
    // issued two MFMA pairs ahead of their use, so the ~50-cycle LDS
    // latency hides under the matrix pipe instead of serializing
    // read -> waitcnt -> MFMA per fragment (the hipcc default here)
    bf16x8 w0[NT], w1[NT];
#pragma unroll
    for (int t = 0; t < NT; ++t) {
      w0[t] = qsa_gb_cvt(q[t].x, q[t].y);     // k0 .. +32
      w1[t] = qsa_gb_cvt(q[t].z, q[t].w);     // k0+32 .. +64
    }
    const unsigned short* abase = &atile[cur][col * QSA_GB_PAD + hi * 8];
#define QSA_GB_RD0(m) \
  (*reinterpret_cast<const bf16x8*>(abase + (m) * 16 * QSA_GB_PAD))
#define QSA_GB_RD1(m) \
  (*reinterpret_cast<const bf16x8*>(abase + (m) * 16 * QSA_GB_PAD + 32))
    bf16x8 p0[3], p1[3];
    p0[0] = QSA_GB_RD0(0);
    p1[0] = QSA_GB_RD1(0);
    if (M_FRAGS > 1) {
      p0[1] = QSA_GB_RD0(1);
      p1[1] = QSA_GB_RD1(1);
    }
#pragma unroll
    for (int m = 0; m < M_FRAGS; ++m) {
      if (m + 2 < M_FRAGS) {
        p0[(m + 2) % 3] = QSA_GB_RD0(m + 2);
        p1[(m + 2) % 3] = QSA_GB_RD1(m + 2);
      }
#pragma unroll
      for (int t = 0; t < NT; ++t) {
        acc[t][m] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
            p0[m % 3], w0[t], acc[t][m], 0, 0, 0);
        acc[t][m] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
            p1[m % 3], w1[t], acc[t][m], 0, 0, 0);
      }
    }
#undef QSA_GB_RD0
#undef QSA_GB_RD1
    // ---- drain staged loads into the other buffer ---------------------
    if (have_next) {
#pragma unroll
      for (int i = 0; i < PER_T; ++i) {
        const int row = frow + i * 32;
        *reinterpret_cast<bf16x8*>(
            &atile[cur ^ 1][row * QSA_GB_PAD + fc8]) = stage[i];
      }
#pragma unroll
      for (int t = 0; t < NT; ++t) q[t] = q_next[t];
    }
    __syncthreads();
  }

  // ---- epilogue: C tile row = m*16 + hi*4 + r, col = lane&15 ----------
#pragma unroll
  for (int t = 0; t < NT; ++t) {
    const int ncol = (nt16 + t) * 16 + col;
    if (SPLITK == 1) {
      const float s = scale[ncol];
#pragma unroll
      for (int m = 0; m < M_FRAGS; ++m) {
#pragma unroll
        for (int r = 0; r < 4; ++r) {
          const int row = m * 16 + hi * 4 + r;
          if (row < M)
            Cbf[(long long)row * N + ncol] = f32_to_bf16(acc[t][m][r] * s);
        }
      }
    } else {
      float* slab = ws + (long long)blockIdx.y * M * N;
#pragma unroll
      for (int m = 0; m < M_FRAGS; ++m) {
#pragma unroll
        for (int r = 0; r < 4; ++r) {
          const int row = m * 16 + hi * 4 + r;
          if (row < M) slab[(long long)row * N + ncol] = acc[t][m][r];
        }
      }
    }
  }
}

__global__ void qsa_splitk_reduce(const float* __restrict__ ws,
                                  const float* __restrict__ scale,
                                  unsigned short* __restrict__ Cbf,
                                  long long mn, int N, int splitk) {
  const long long i = (long long)blockIdx.x * blockDim.x + threadIdx.x;
  if (i >= mn) return;
  float v = 0.f;
  for (int s = 0; s < splitk; ++s) v += ws[(long long)s * mn + i];
  Cbf[i] = f32_to_bf16(v * scale[i % N]);
}

extern "C" void qsa_gemm_fp8_batch_launch(
    const unsigned short* A, const unsigned char* Qf, const float* scale,
    unsigned short* Cbf, float* ws, int M, int N, long long K,
    long long lda, int splitk, hipStream_t stream) {
  const u32x4* Q = reinterpret_cast<const u32x4*>(Qf);
  const int mf = (M + 15) / 16;
  const int nt = (N % 128 == 0 && mf <= 12) ? 2 : 1;
#define QSA_CASE(MF, SK, NTT)                                             \
  if (mf <= MF && splitk == SK && nt == NTT) {                            \
    hipLaunchKernelGGL((qsa_gemm_fp8_batch<MF, SK, NTT>),                 \
                       dim3(N / (64 * NTT), SK), dim3(256), 0, stream,    \
                       A, Q, scale, Cbf, ws, M, N, K, lda);               \
    goto reduce;                                                          \
  }
  QSA_CASE(4, 1, 2) QSA_CASE(8, 1, 2) QSA_CASE(12, 1, 2)
  QSA_CASE(4, 2, 2) QSA_CASE(8, 2, 2) QSA_CASE(12, 2, 2)
  QSA_CASE(4, 4, 2) QSA_CASE(8, 4, 2) QSA_CASE(12, 4, 2)
  QSA_CASE(4, 1, 1) QSA_CASE(8, 1, 1) QSA_CASE(12, 1, 1) QSA_CASE(16, 1, 1)
  QSA_CASE(4, 2, 1) QSA_CASE(8, 2, 1) QSA_CASE(12, 2, 1) QSA_CASE(16, 2, 1)
  QSA_CASE(4, 4, 1) QSA_CASE(8, 4, 1) QSA_CASE(12, 4, 1) QSA_CASE(16, 4, 1)
#undef QSA_CASE
  return;  // unsupported combination (host validates)
reduce:
  if (splitk > 1) {
    const long long mn = (long long)M * N;
    hipLaunchKernelGGL(qsa_splitk_reduce,
                       dim3((unsigned)((mn + 255) / 256)), dim3(256), 0,
                       stream, ws, scale, Cbf, mn, N, splitk);
  }
}
