// FP8 (OCP e4m3) weight-streaming decode GEMM for CDNA4 (gfx950):
// C[M,N] = A[M,K] @ (s[N] * Q[N,K])^T, M <= 32, A bf16, Q fp8, C bf16.
// Serves the K4 agent-LLM decode projections.
//
// The bf16 skinny kernel (skinny_gemm.hip) is HBM-bound on the WEIGHT
// stream (PMC: ~3% MFMA util); its ceiling is W bytes / 6.3 TB/s.  FP8
// weights HALVE that stream: per-output-channel symmetric quantization
// (s[n] = max|W[n,:]| / 448, the standard W8A16 weight-only scheme) keeps
// activations bf16 and applies the scale once in the epilogue — zero
// inner-loop cost.  Dequant in-kernel:
//   v_cvt_pk_f32_fp8 (2 fp8 -> 2 f32, exact)
//   v_perm_b32       (pack the two f32 high halves -> 2 bf16)
// The f32 -> bf16 TRUNCATION is exact: e4m3 has 3 mantissa bits, bf16
// has 8, and every e4m3 value (normals and denormals) is exactly
// representable in bf16 — the dequantized operand is bit-exact, so the
// MFMA math matches a bf16 kernel running on the dequantized weights.
//
// Pack layout [N/16, K/64, 16, 4, 2, 8] ("fragment-pair-major"): one lane
// reads ONE 16-byte vector covering its 8 fp8 for TWO consecutive
// 16x16x32 k-fragments -> full-width dwordx4 loads on a byte stream.
// Gfx950 also has native fp8 MFMA, but using it would force the
// ACTIVATIONS to fp8 too (non-scaled fp8 MFMA is A8W8); this kernel is
// bandwidth-bound, so bf16 MFMA at half the weight bytes is the same
// speed with better numerics.
#include "common.h"

using bf16x8 = __attribute__((ext_vector_type(8))) short;
using f32x2v = __attribute__((ext_vector_type(2))) float;
using f32x4 = __attribute__((ext_vector_type(4))) float;
using u32x4 = __attribute__((ext_vector_type(4))) unsigned int;

#define QSA_KCH8 256  // k per wave-iteration: 4 loads, 8 MFMA k-steps

// 8 fp8 bytes (as 2 dwords) -> bf16x8 fragment, exact.
__device__ __forceinline__ bf16x8 fp8x8_to_bf16x8(unsigned int a,
                                                  unsigned int b) {
  f32x2v f01 = __builtin_amdgcn_cvt_pk_f32_fp8(a, false);
  f32x2v f23 = __builtin_amdgcn_cvt_pk_f32_fp8(a, true);
  f32x2v f45 = __builtin_amdgcn_cvt_pk_f32_fp8(b, false);
  f32x2v f67 = __builtin_amdgcn_cvt_pk_f32_fp8(b, true);
  union { unsigned int u; float f; } u0, u1;
  unsigned int p[4];
  u0.f = f01.x; u1.f = f01.y;
  p[0] = __builtin_amdgcn_perm(u1.u, u0.u, 0x07060302u);
  u0.f = f23.x; u1.f = f23.y;
  p[1] = __builtin_amdgcn_perm(u1.u, u0.u, 0x07060302u);
  u0.f = f45.x; u1.f = f45.y;
  p[2] = __builtin_amdgcn_perm(u1.u, u0.u, 0x07060302u);
  u0.f = f67.x; u1.f = f67.y;
  p[3] = __builtin_amdgcn_perm(u1.u, u0.u, 0x07060302u);
  union { unsigned int u[4]; bf16x8 v; } out;
  out.u[0] = p[0]; out.u[1] = p[1]; out.u[2] = p[2]; out.u[3] = p[3];
  return out.v;
}

// TILES n-tiles (16 cols each) per workgroup: the A (activation) loads —
// L2 traffic that rivals the fp8 W stream at 16 cols/WG — are loaded once
// per k-step and reused across all TILES MFMA streams.  TILES=4 for the
// big-N projections (wgu/lm_head), 1 for the small ones (grid must still
// exceed 256 CUs).
template <int WAVES, int TILES, bool NT>
__global__ void __launch_bounds__(WAVES * 64)
qsa_skinny_gemm_fp8_t(const unsigned short* __restrict__ A,  // [M,K] bf16
                      const u32x4* __restrict__ Qf,   // packed fp8 stream
                      const float* __restrict__ scale,     // [N] f32
                      unsigned short* __restrict__ Cbf,    // [M, N]
                      int M, int N, long long K, long long lda) {
  const int nt0 = blockIdx.x * TILES;     // first 16-col n-tile
  const int wave = threadIdx.x >> 6;
  const int lane = threadIdx.x & 63;
  const long long kchunks = K / QSA_KCH8;  // K % 256 == 0

  f32x4 acc0[TILES], acc1[TILES];
#pragma unroll
  for (int t = 0; t < TILES; ++t) {
    acc0[t] = {0.f, 0.f, 0.f, 0.f};
    acc1[t] = {0.f, 0.f, 0.f, 0.f};
  }

  // A rows clamp out-of-batch to M-1 (store is guarded; see bf16 kernel)
  const int arow = lane & 15;
  const int akoff = (lane >> 4) * 8;
  const int r0 = min(arow, M - 1);
  const int r1 = min(16 + arow, M - 1);
  const unsigned short* a0base = A + (long long)r0 * lda + akoff;
  const unsigned short* a1base = A + (long long)r1 * lda + akoff;

  // Q stream: 1024-B block per (nt, kb=64k); this lane's 16 B at byte
  // (lane&15)*64 + (lane>>4)*16 inside the block (uint4 units below).
  const u32x4* qbase[TILES];
#pragma unroll
  for (int t = 0; t < TILES; ++t)
    qbase[t] = Qf + ((long long)(nt0 + t) * (K >> 6)) * 64 +
               (long long)((lane & 15) * 4 + (lane >> 4));

  for (long long c = wave; c < kchunks; c += WAVES) {
    const long long k0 = c * QSA_KCH8;
#pragma unroll
    for (int s = 0; s < 4; ++s) {     // 4 k-steps x 64 k = 256 k
      const long long kb = (k0 >> 6) + s;
      u32x4 q[TILES];
#pragma unroll
      for (int t = 0; t < TILES; ++t) {
        if (NT) {
          q[t] = __builtin_nontemporal_load(qbase[t] + kb * 64);
        } else {
          q[t] = *reinterpret_cast<const u32x4*>(
              __builtin_assume_aligned(qbase[t] + kb * 64, 16));
        }
      }
      const long long ak = k0 + s * 64;
      const bf16x8 a00 = *reinterpret_cast<const bf16x8*>(a0base + ak);
      const bf16x8 a10 = *reinterpret_cast<const bf16x8*>(a1base + ak);
      const bf16x8 a01 = *reinterpret_cast<const bf16x8*>(a0base + ak + 32);
      const bf16x8 a11 = *reinterpret_cast<const bf16x8*>(a1base + ak + 32);
#pragma unroll
      for (int t = 0; t < TILES; ++t) {
        const bf16x8 w0 = fp8x8_to_bf16x8(q[t].x, q[t].y);   // k .. +32
        const bf16x8 w1 = fp8x8_to_bf16x8(q[t].z, q[t].w);   // +32 .. +64
        acc0[t] =
            __builtin_amdgcn_mfma_f32_16x16x32_bf16(a00, w0, acc0[t], 0, 0, 0);
        acc1[t] =
            __builtin_amdgcn_mfma_f32_16x16x32_bf16(a10, w0, acc1[t], 0, 0, 0);
        acc0[t] =
            __builtin_amdgcn_mfma_f32_16x16x32_bf16(a01, w1, acc0[t], 0, 0, 0);
        acc1[t] =
            __builtin_amdgcn_mfma_f32_16x16x32_bf16(a11, w1, acc1[t], 0, 0, 0);
      }
    }
  }

  // ---- cross-wave K-reduction in LDS + per-channel scale ---------------
  __shared__ float red[WAVES][64][8];
#pragma unroll
  for (int t = 0; t < TILES; ++t) {
    if (t > 0) __syncthreads();   // reuse the LDS slab per tile
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      red[wave][lane][r] = acc0[t][r];
      red[wave][lane][4 + r] = acc1[t][r];
    }
    __syncthreads();
    if (wave == 0) {
      const int ncol = (nt0 + t) * 16 + (lane & 15);
      const int mrow = (lane >> 4) * 4;
      const float s = scale[ncol];
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        float v0 = 0.f, v1 = 0.f;
#pragma unroll
        for (int wv = 0; wv < WAVES; ++wv) {
          v0 += red[wv][lane][r];
          v1 += red[wv][lane][4 + r];
        }
        const int m0 = mrow + r;
        if (m0 < M) Cbf[(long long)m0 * N + ncol] = f32_to_bf16(v0 * s);
        if (16 + m0 < M)
          Cbf[(long long)(16 + m0) * N + ncol] = f32_to_bf16(v1 * s);
      }
    }
  }
}

extern "C" void qsa_skinny_gemm_fp8_launch(
    const unsigned short* A, const unsigned char* Qf, const float* scale,
    unsigned short* Cbf, int M, int N, long long K, long long lda,
    hipStream_t stream) {
  // Per-shape winners from the measured sweep on MI355X
  // (profiles/fp8_decode_gemm.md): huge N wants more tiles + fewer
  // waves (lm_head w2t8 = 4.6 TB/s fp8 bytes), mid N wants t4, qkv-size
  // t2, small/deep-K t1.
  const u32x4* Q = reinterpret_cast<const u32x4*>(Qf);
  if (N % 128 == 0 && N >= 65536) {          // lm_head class
    hipLaunchKernelGGL((qsa_skinny_gemm_fp8_t<2, 8, false>), dim3(N / 128),
                       dim3(128), 0, stream, A, Q, scale, Cbf, M, N, K,
                       lda);
  } else if (N % 64 == 0 && N >= 16384) {    // wgu class
    hipLaunchKernelGGL((qsa_skinny_gemm_fp8_t<8, 4, false>), dim3(N / 64),
                       dim3(512), 0, stream, A, Q, scale, Cbf, M, N, K,
                       lda);
  } else if (N % 32 == 0 && N >= 6144 && K < 8192) {   // qkv class
    hipLaunchKernelGGL((qsa_skinny_gemm_fp8_t<8, 2, false>), dim3(N / 32),
                       dim3(512), 0, stream, A, Q, scale, Cbf, M, N, K,
                       lda);
  } else {                                   // wo / wdown class
    hipLaunchKernelGGL((qsa_skinny_gemm_fp8_t<8, 1, false>), dim3(N / 16),
                       dim3(512), 0, stream, A, Q, scale, Cbf, M, N, K,
                       lda);
  }
}

extern "C" void qsa_skinny_gemm_fp8_probe_launch(
    const unsigned short* A, const unsigned char* Qf, const float* scale,
    unsigned short* Cbf, int M, int N, long long K, long long lda,
    int waves, int tiles, int nt, hipStream_t stream) {
#define QSA_CASE(W, T, NTB)                                               \
  if (waves == W && tiles == T && nt == (int)NTB) {                       \
    hipLaunchKernelGGL((qsa_skinny_gemm_fp8_t<W, T, NTB>),                \
                       dim3(N / (16 * T)), dim3(W * 64), 0, stream, A,    \
                       reinterpret_cast<const u32x4*>(Qf), scale, Cbf,    \
                       M, N, K, lda);                                     \
    return;                                                               \
  }
  QSA_CASE(8, 1, false) QSA_CASE(8, 1, true) QSA_CASE(8, 2, false)
  QSA_CASE(8, 4, false) QSA_CASE(8, 4, true) QSA_CASE(8, 8, false)
  QSA_CASE(16, 1, false) QSA_CASE(16, 2, false) QSA_CASE(16, 4, false)
  QSA_CASE(4, 4, false) QSA_CASE(4, 8, false) QSA_CASE(2, 8, false)
#undef QSA_CASE
}
