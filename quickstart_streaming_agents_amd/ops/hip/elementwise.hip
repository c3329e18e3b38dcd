// Fused elementwise / normalization kernels for the MI355X decode path.
// All bf16 I/O is vectorized as uint4 (8 bf16 = 16 B per lane) per the
// CDNA4 guide (scalar bf16 loads are ~2-2.5x slower).
#include "common.h"

// ---------------------------------------------------------------------------
// RMSNorm: y = x / sqrt(mean(x^2) + eps) * w      (rows x H, H % 8 == 0)
// Optional fused residual: res = x + res first; y = rmsnorm(res) * w.
// Replaces the reference's managed-LLM internals with an on-GPU op (K4).
// ---------------------------------------------------------------------------
__global__ void __launch_bounds__(256)
qsa_rmsnorm_kernel(const unsigned short* __restrict__ x,
                   const unsigned short* __restrict__ w,
                   unsigned short* __restrict__ y,
                   unsigned short* __restrict__ res,  // nullptr = no residual
                   int H, float eps) {
  const long long row = blockIdx.x;
  const unsigned short* xr = x + row * H;
  unsigned short* yr = y + row * H;
  unsigned short* rr = res ? res + row * H : nullptr;
  __shared__ float scratch[8];

  // register-cache the row between the two passes (H <= 8192 with 256
  // threads = <=4 x uint4 per thread; compile-time indices so the cache
  // stays in VGPRs, guide rule 20) — the normalize pass then re-reads
  // nothing, which halves the dependent-latency chain of this small-grid
  // decode kernel.
  uint4 buf[4];
  float ss = 0.f;
#pragma unroll
  for (int c = 0; c < 4; ++c) {
    const int i = (threadIdx.x + c * blockDim.x) * 8;
    if (i >= H) break;
    uint4 v = *reinterpret_cast<const uint4*>(xr + i);
    float acc = 0.f;
    unsigned int pk[4] = {v.x, v.y, v.z, v.w};
    if (rr) {
      uint4 rv = *reinterpret_cast<const uint4*>(rr + i);
      unsigned int rpk[4] = {rv.x, rv.y, rv.z, rv.w};
      uint4 outv;
      unsigned int* po = &outv.x;
#pragma unroll
      for (int j = 0; j < 4; ++j) {
        float2 a = bf16x2_to_f32x2(pk[j]);
        float2 b = bf16x2_to_f32x2(rpk[j]);
        float s0 = a.x + b.x, s1 = a.y + b.y;
        acc = fmaf(s0, s0, acc);
        acc = fmaf(s1, s1, acc);
        po[j] = f32x2_to_bf16x2(s0, s1);
      }
      *reinterpret_cast<uint4*>(rr + i) = outv;  // updated residual stream
      buf[c] = outv;
    } else {
#pragma unroll
      for (int j = 0; j < 4; ++j) {
        float2 a = bf16x2_to_f32x2(pk[j]);
        acc = fmaf(a.x, a.x, acc);
        acc = fmaf(a.y, a.y, acc);
      }
      buf[c] = v;
    }
    ss += acc;
  }
  // rows wider than the register cache (H > 8192): uncached tail does the
  // full pass-1 work (residual add + write) and pass 2 re-reads
  const bool cached = H <= (int)blockDim.x * 8 * 4;
  if (!cached) {
    for (int i = (threadIdx.x + 4 * blockDim.x) * 8; i < H;
         i += blockDim.x * 8) {
      uint4 v = *reinterpret_cast<const uint4*>(xr + i);
      unsigned int pk[4] = {v.x, v.y, v.z, v.w};
      if (rr) {
        uint4 rv = *reinterpret_cast<const uint4*>(rr + i);
        unsigned int rpk[4] = {rv.x, rv.y, rv.z, rv.w};
        uint4 outv;
        unsigned int* po = &outv.x;
#pragma unroll
        for (int j = 0; j < 4; ++j) {
          float2 a = bf16x2_to_f32x2(pk[j]);
          float2 b = bf16x2_to_f32x2(rpk[j]);
          float s0 = a.x + b.x, s1 = a.y + b.y;
          ss = fmaf(s0, s0, ss);
          ss = fmaf(s1, s1, ss);
          po[j] = f32x2_to_bf16x2(s0, s1);
        }
        *reinterpret_cast<uint4*>(rr + i) = outv;
      } else {
#pragma unroll
        for (int j = 0; j < 4; ++j) {
          float2 a = bf16x2_to_f32x2(pk[j]);
          ss = fmaf(a.x, a.x, ss);
          ss = fmaf(a.y, a.y, ss);
        }
      }
    }
  }
  ss = block_reduce_sum(ss, scratch);
  const float inv = rsqrtf(ss / (float)H + eps);

#pragma unroll
  for (int c = 0; c < 4; ++c) {
    const int i = (threadIdx.x + c * blockDim.x) * 8;
    if (i >= H) break;
    uint4 v = buf[c];
    uint4 wv = *reinterpret_cast<const uint4*>(w + i);
    unsigned int pk[4] = {v.x, v.y, v.z, v.w};
    unsigned int wk[4] = {wv.x, wv.y, wv.z, wv.w};
    uint4 outv;
    unsigned int* po = &outv.x;
#pragma unroll
    for (int j = 0; j < 4; ++j) {
      float2 a = bf16x2_to_f32x2(pk[j]);
      float2 b = bf16x2_to_f32x2(wk[j]);
      po[j] = f32x2_to_bf16x2(a.x * inv * b.x, a.y * inv * b.y);
    }
    *reinterpret_cast<uint4*>(yr + i) = outv;
  }
  if (!cached) {
    const unsigned short* src0 = rr ? rr : xr;
    for (int i = (threadIdx.x + 4 * blockDim.x) * 8; i < H;
         i += blockDim.x * 8) {
      uint4 v = *reinterpret_cast<const uint4*>(src0 + i);
      uint4 wv = *reinterpret_cast<const uint4*>(w + i);
      unsigned int pk[4] = {v.x, v.y, v.z, v.w};
      unsigned int wk[4] = {wv.x, wv.y, wv.z, wv.w};
      uint4 outv;
      unsigned int* po = &outv.x;
#pragma unroll
      for (int j = 0; j < 4; ++j) {
        float2 a = bf16x2_to_f32x2(pk[j]);
        float2 b = bf16x2_to_f32x2(wk[j]);
        po[j] = f32x2_to_bf16x2(a.x * inv * b.x, a.y * inv * b.y);
      }
      *reinterpret_cast<uint4*>(yr + i) = outv;
    }
  }
}

// ---------------------------------------------------------------------------
// SwiGLU: y = silu(gate) * up   over [n] bf16 (flat), 8 elems per lane.
// ---------------------------------------------------------------------------
__global__ void __launch_bounds__(256)
qsa_swiglu_kernel(const unsigned short* __restrict__ gate,
                  const unsigned short* __restrict__ up,
                  unsigned short* __restrict__ y, long long rows,
                  long long cols, long long in_stride) {
  long long i = ((long long)blockIdx.x * blockDim.x + threadIdx.x) * 8;
  const long long stride = (long long)gridDim.x * blockDim.x * 8;
  const long long n = rows * cols;
  for (; i + 7 < n; i += stride) {
    const long long r = i / cols, cidx = i % cols;
    const long long off = r * in_stride + cidx;
    uint4 g = *reinterpret_cast<const uint4*>(gate + off);
    uint4 u = *reinterpret_cast<const uint4*>(up + off);
    unsigned int gp[4] = {g.x, g.y, g.z, g.w};
    unsigned int upk[4] = {u.x, u.y, u.z, u.w};
    uint4 outv;
    unsigned int* po = &outv.x;
#pragma unroll
    for (int j = 0; j < 4; ++j) {
      float2 a = bf16x2_to_f32x2(gp[j]);
      float2 b = bf16x2_to_f32x2(upk[j]);
      float s0 = a.x / (1.f + __expf(-a.x));
      float s1 = a.y / (1.f + __expf(-a.y));
      po[j] = f32x2_to_bf16x2(s0 * b.x, s1 * b.y);
    }
    *reinterpret_cast<uint4*>(y + i) = outv;
  }
}

// ---------------------------------------------------------------------------
// RoPE (llama-style half-rotation), in-place on q [B, QH, D] and k [B, KVH, D]
// cos/sin table [max_pos, D/2] f32; positions [B].
// One block of D/2 threads per (b, head); heads 0..QH-1 are q, rest are k.
// ---------------------------------------------------------------------------
__global__ void
qsa_rope_kernel(unsigned short* __restrict__ q, unsigned short* __restrict__ k,
                const float* __restrict__ cos_t, const float* __restrict__ sin_t,
                const int* __restrict__ pos, int B, int QH, int KVH, int D,
                long long qstride, long long kstride) {
  const int bh = blockIdx.x;
  const int nheads = QH + KVH;
  const int b = bh / nheads;
  const int h = bh % nheads;
  const int half = D / 2;
  const int d = threadIdx.x;
  if (d >= half) return;
  unsigned short* base =
      (h < QH) ? q + (long long)b * qstride + (long long)h * D
               : k + (long long)b * kstride + (long long)(h - QH) * D;
  const long long toff = (long long)pos[b] * half + d;
  const float c = cos_t[toff], s = sin_t[toff];
  const float x0 = bf16_to_f32(base[d]);
  const float x1 = bf16_to_f32(base[d + half]);
  base[d] = f32_to_bf16(x0 * c - x1 * s);
  base[d + half] = f32_to_bf16(x0 * s + x1 * c);
}

// ---------------------------------------------------------------------------
// Row softmax (f32), optionally causal within a [rows, cols] score block:
// mask col > row + col_offset.  Used by the chunked prefill attention.
// ---------------------------------------------------------------------------
__global__ void __launch_bounds__(256)
qsa_softmax_rows_kernel(float* __restrict__ scores, int rows, int cols,
                        int col_offset, int causal, int row_mod,
                        const int* __restrict__ row_limits) {
  const int row = blockIdx.x;
  float* r = scores + (long long)row * cols;
  int limit = cols;
  if (causal) {
    const int pos = row_mod > 0 ? row % row_mod : row;
    limit = min(cols, pos + col_offset + 1);
  }
  if (row_limits) limit = min(limit, row_limits[row]);
  __shared__ float scratch[8];

  float mx = -3.0e38f;
  for (int i = threadIdx.x; i < limit; i += blockDim.x)
    mx = fmaxf(mx, r[i]);
  // block max via LDS
  {
    const int lane = threadIdx.x & 63, wave = threadIdx.x >> 6;
    float wm = wave_reduce_max(mx);
    if (lane == 0) scratch[wave] = wm;
    __syncthreads();
    float m = -3.0e38f;
    for (int i = 0; i < (int)(blockDim.x >> 6); ++i) m = fmaxf(m, scratch[i]);
    __syncthreads();
    mx = m;
  }
  float sum = 0.f;
  for (int i = threadIdx.x; i < limit; i += blockDim.x) {
    float e = __expf(r[i] - mx);
    r[i] = e;
    sum += e;
  }
  sum = block_reduce_sum(sum, scratch);
  const float inv = (sum > 0.f) ? 1.f / sum : 0.f;
  for (int i = threadIdx.x; i < cols; i += blockDim.x)
    r[i] = (i < limit) ? r[i] * inv : 0.f;
}

// ---------------------------------------------------------------------------
// C launchers (ops.cpp links against these; no RDC needed)
// ---------------------------------------------------------------------------
extern "C" void qsa_rmsnorm_launch(const unsigned short* x,
                                   const unsigned short* w, unsigned short* y,
                                   unsigned short* res, long long rows, int H,
                                   float eps, hipStream_t stream) {
  hipLaunchKernelGGL(qsa_rmsnorm_kernel, dim3(rows), dim3(256), 0, stream, x,
                     w, y, res, H, eps);
}

extern "C" void qsa_swiglu_launch(const unsigned short* gate,
                                  const unsigned short* up, unsigned short* y,
                                  long long rows, long long cols,
                                  long long in_stride, int blocks,
                                  hipStream_t stream) {
  hipLaunchKernelGGL(qsa_swiglu_kernel, dim3(blocks), dim3(256), 0, stream,
                     gate, up, y, rows, cols, in_stride);
}

extern "C" void qsa_rope_launch(unsigned short* q, unsigned short* k,
                                const float* cos_t, const float* sin_t,
                                const int* pos, int B, int QH, int KVH, int D,
                                long long qstride, long long kstride,
                                hipStream_t stream) {
  hipLaunchKernelGGL(qsa_rope_kernel, dim3(B * (QH + KVH)), dim3(D / 2), 0,
                     stream, q, k, cos_t, sin_t, pos, B, QH, KVH, D, qstride,
                     kstride);
}

extern "C" void qsa_softmax_rows_launch(float* scores, int rows, int cols,
                                        int col_offset, int causal,
                                        int row_mod, const int* row_limits,
                                        hipStream_t stream) {
  hipLaunchKernelGGL(qsa_softmax_rows_kernel, dim3(rows), dim3(256), 0, stream,
                     scores, rows, cols, col_offset, causal, row_mod,
                     row_limits);
}

// ---------------------------------------------------------------------------
// bf16 masked row softmax with folded scale (padded-batch prefill attention):
// in-place on bf16 scores [rows, cols]; per-row valid length from
// row_limits; f32 max/sum internally; beyond-limit columns zeroed so the
// downstream PV bmm sees exact ragged/causal masking.  Vectorized 8-wide
// (uint4 = 8 bf16) — the f32 variant's 3 passes over an f32 matrix plus
// the cast kernels around it were ~25% of prefill GPU time.
// ---------------------------------------------------------------------------
__global__ void __launch_bounds__(256)
qsa_softmax_rows_bf16_kernel(unsigned short* __restrict__ scores, int cols,
                             float scale,
                             const int* __restrict__ row_limits) {
  const long long row = blockIdx.x;
  unsigned short* r = scores + row * cols;
  const int limit = min(row_limits[row], cols);
  __shared__ float scratch[8];
  const int lane = threadIdx.x & 63, wave = threadIdx.x >> 6;
  const int nw = blockDim.x >> 6;

  float mx = -3.0e38f;
  for (int i = threadIdx.x * 8; i < limit; i += blockDim.x * 8) {
    uint4 v4 = *reinterpret_cast<const uint4*>(r + i);
    const unsigned int vv[4] = {v4.x, v4.y, v4.z, v4.w};
#pragma unroll
    for (int j = 0; j < 4; ++j) {
      float2 f = bf16x2_to_f32x2(vv[j]);
      if (i + 2 * j < limit) mx = fmaxf(mx, f.x);
      if (i + 2 * j + 1 < limit) mx = fmaxf(mx, f.y);
    }
  }
  {
    float wm = wave_reduce_max(mx);
    if (lane == 0) scratch[wave] = wm;
    __syncthreads();
    float m = -3.0e38f;
    for (int i = 0; i < nw; ++i) m = fmaxf(m, scratch[i]);
    __syncthreads();
    mx = m * scale;
  }
  // exp pass into registers is wasteful at this width; do exp+sum, then
  // a normalize pass re-computing exp (still half the f32 variant's bytes)
  float sum = 0.f;
  for (int i = threadIdx.x * 8; i < limit; i += blockDim.x * 8) {
    uint4 v4 = *reinterpret_cast<const uint4*>(r + i);
    const unsigned int vv[4] = {v4.x, v4.y, v4.z, v4.w};
#pragma unroll
    for (int j = 0; j < 4; ++j) {
      float2 f = bf16x2_to_f32x2(vv[j]);
      if (i + 2 * j < limit) sum += __expf(f.x * scale - mx);
      if (i + 2 * j + 1 < limit) sum += __expf(f.y * scale - mx);
    }
  }
  sum = block_reduce_sum(sum, scratch);
  const float inv = (sum > 0.f) ? 1.f / sum : 0.f;
  const int cols8 = (cols + 7) & ~7;
  for (int i = threadIdx.x * 8; i < cols8; i += blockDim.x * 8) {
    uint4 v4 = *reinterpret_cast<const uint4*>(r + i);
    unsigned int vv[4] = {v4.x, v4.y, v4.z, v4.w};
    uint4 o4;
    unsigned int* oo = reinterpret_cast<unsigned int*>(&o4);
#pragma unroll
    for (int j = 0; j < 4; ++j) {
      float2 f = bf16x2_to_f32x2(vv[j]);
      const float a = (i + 2 * j < limit) ? __expf(f.x * scale - mx) * inv
                                          : 0.f;
      const float b = (i + 2 * j + 1 < limit)
                          ? __expf(f.y * scale - mx) * inv : 0.f;
      oo[j] = f32x2_to_bf16x2(a, b);
    }
    *reinterpret_cast<uint4*>(r + i) = o4;
  }
}

extern "C" void qsa_softmax_rows_bf16_launch(unsigned short* scores,
                                             long long rows, int cols,
                                             float scale,
                                             const int* row_limits,
                                             hipStream_t stream) {
  hipLaunchKernelGGL(qsa_softmax_rows_bf16_kernel, dim3((unsigned)rows),
                     dim3(256), 0, stream, scores, cols, scale, row_limits);
}
