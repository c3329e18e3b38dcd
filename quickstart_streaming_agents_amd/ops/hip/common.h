// Shared device helpers for the qsa MI355X (gfx950) kernels.
// Wave width is 64 on CDNA4 — hard-coded per the platform guide.
#pragma once

#include <hip/hip_runtime.h>

#define QSA_WAVE 64

// ---- bf16 <-> f32 (manual, branch-free RNE) -------------------------------
__device__ __forceinline__ float bf16_to_f32(unsigned short h) {
  union { unsigned int u; float f; } v;
  v.u = ((unsigned int)h) << 16;
  return v.f;
}

__device__ __forceinline__ unsigned short f32_to_bf16(float f) {
  union { unsigned int u; float f; } v;
  v.f = f;
  // round-to-nearest-even; NaN-safe (quiet the NaN)
  unsigned int u = v.u;
  if ((u & 0x7fffffffu) > 0x7f800000u) return (unsigned short)((u >> 16) | 0x40);
  unsigned int rounding = 0x7fffu + ((u >> 16) & 1u);
  return (unsigned short)((u + rounding) >> 16);
}

// unpack 2 bf16 held in one uint (lo, hi)
__device__ __forceinline__ float2 bf16x2_to_f32x2(unsigned int p) {
  union { unsigned int u; float f; } lo, hi;
  lo.u = p << 16;
  hi.u = p & 0xffff0000u;
  return make_float2(lo.f, hi.f);
}

__device__ __forceinline__ unsigned int f32x2_to_bf16x2(float a, float b) {
  return (unsigned int)f32_to_bf16(a) | ((unsigned int)f32_to_bf16(b) << 16);
}

// ---- wave reductions (64-lane) --------------------------------------------
__device__ __forceinline__ float wave_reduce_sum(float v) {
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) v += __shfl_xor(v, off, QSA_WAVE);
  return v;
}

__device__ __forceinline__ float wave_reduce_max(float v) {
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) v = fmaxf(v, __shfl_xor(v, off, QSA_WAVE));
  return v;
}

// Block reduction via LDS (callers pass scratch of >= nwaves floats).
// All threads receive the result.
__device__ __forceinline__ float block_reduce_sum(float v, float* scratch) {
  const int lane = threadIdx.x & (QSA_WAVE - 1);
  const int wave = threadIdx.x / QSA_WAVE;
  const int nwaves = blockDim.x / QSA_WAVE;
  v = wave_reduce_sum(v);
  if (lane == 0) scratch[wave] = v;
  __syncthreads();
  float total = 0.f;
#pragma unroll 4
  for (int i = 0; i < nwaves; ++i) total += scratch[i];
  __syncthreads();
  return total;
}

#define QSA_CHECK(cond, msg)                                            \
  TORCH_CHECK(cond, "qsa_hip: ", msg)

typedef __attribute__((ext_vector_type(2))) __bf16 qsa_bf16x2_t;
__device__ __forceinline__ qsa_bf16x2_t as_bf16x2(unsigned int u) {
  union { unsigned int u; qsa_bf16x2_t v; } c;
  c.u = u;
  return c.v;
}
