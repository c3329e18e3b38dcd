// Paged-attention DECODE kernel for CDNA4 (gfx950) — the K4 hot op.
//
// Replaces the reference's managed-LLM call (ML_PREDICT 'llm_textgen_model',
// terraform/core/main.tf:461) with on-GPU batched decode attention over a
// paged KV cache resident in 288 GB HBM3E.
//
// Design (one workgroup per (batch b, kv head)):
//   - 4 waves; wave w owns GQA query head r = w, w+4, ... (R = QH/KVH).
//     All waves walk the same pages in the same order, so the 4x K/V re-read
//     is served by the CU's L1/L2 (page K slab = 16 KB, V = 16 KB).
//   - q for the wave's head lives packed bf16x2 in 64 VGPRs (no LDS traffic
//     in the score loop).
//   - K cache layout [page, kvh, D/8, PAGE=64, 8] ("d-major x8"): for a
//     fixed d-slice, lane p reads positions p contiguously -> one fully
//     coalesced 1 KiB wave transaction per 16 B slice.
//   - V cache layout [page, kvh, PAGE, D] (position-major): in the PV pass
//     lane l owns output dims (2l, 2l+1); reading V[pos, 2l..2l+1] is a
//     coalesced 256 B row per position.
//   - online softmax per wave; probs broadcast lane->lane via __shfl
//     (no LDS); each wave writes its head's output row independently.
//
// Every KV byte is read once per wave, coalesced — the kernel targets the
// HBM roofline, which is what decode attention is bound by.
#include "common.h"

#define QSA_PAGE 64

template <int D>
__global__ void __launch_bounds__(256)
qsa_paged_attn_decode(const unsigned short* __restrict__ q,   // [B, QH, D] (row stride qstride)
                      const unsigned short* __restrict__ kc,  // [P, KVH, D/8, 64, 8]
                      const unsigned short* __restrict__ vc,  // [P, KVH, 64, D]
                      const int* __restrict__ block_table,    // [B, max_pages]
                      const int* __restrict__ seq_lens,       // [B]
                      unsigned short* __restrict__ out,       // [B, QH, D]
                      float scale, int B, int QH, int KVH, int max_pages,
                      long long qstride) {
  const int b = blockIdx.x / KVH;
  const int kvh = blockIdx.x % KVH;
  const int wave = threadIdx.x / QSA_WAVE;
  const int lane = threadIdx.x % QSA_WAVE;
  const int R = QH / KVH;
  const int seqlen = seq_lens[b];
  if (seqlen <= 0) return;
  const int npages = (seqlen + QSA_PAGE - 1) / QSA_PAGE;
  const int* btab = block_table + (long long)b * max_pages;

  for (int r = wave; r < R; r += 4) {
    const int qh = kvh * R + r;
    // ---- q packed bf16x2 into regs: D/2 uints (64 for D=128) ----
    unsigned int qpk[D / 2];  // D bf16 = D/2 packed uints
    const unsigned int* qsrc = reinterpret_cast<const unsigned int*>(
        q + (long long)b * qstride + (long long)qh * D);
#pragma unroll
    for (int i = 0; i < D / 2; ++i) qpk[i] = qsrc[i];

    // PV decomposition: lane = pg * DG + dg — dim-group dg owns 8 output
    // dims, position-group pg covers PG=64/DG positions per iteration, so a
    // V read is 64 lanes x 16 B = 1 KiB fully coalesced and the page's PV
    // takes PAGE/PG wide iterations instead of 64 scalar ones.
    constexpr int DG = D / 8;        // lanes per output row (16 for D=128)
    constexpr int PG = QSA_WAVE / DG;  // positions per iteration (4)
    const int dg = lane % DG;
    const int pg = lane / DG;

    float m = -3.0e38f, s = 0.f;
    float o8[8];
#pragma unroll
    for (int j = 0; j < 8; ++j) o8[j] = 0.f;

    for (int pi = 0; pi < npages; ++pi) {
      const int page = btab[pi];
      const int pos = pi * QSA_PAGE + lane;
      const bool valid = pos < seqlen;
      // ---- score: dot(q, K[:, pos]) over d-slices of 8 (lane = position)
      float sc = 0.f;
      const uint4* kbase = reinterpret_cast<const uint4*>(
          kc + ((((long long)page * KVH + kvh) * (D / 8)) * QSA_PAGE) * 8);
#pragma unroll
      for (int d0 = 0; d0 < D / 8; ++d0) {
        uint4 kv4 = kbase[d0 * QSA_PAGE + lane];
        unsigned int kk[4] = {kv4.x, kv4.y, kv4.z, kv4.w};
#pragma unroll
        for (int j = 0; j < 4; ++j)   // v_dot2_f32_bf16: 2 MACs/instr
          sc = __builtin_amdgcn_fdot2_f32_bf16(
              as_bf16x2(kk[j]), as_bf16x2(qpk[d0 * 4 + j]), sc, false);
      }
      sc = valid ? sc * scale : -3.0e38f;
      // ---- online softmax over this page ----
      const float pmax = wave_reduce_max(sc);
      const float m_new = fmaxf(m, pmax);
      float alpha = __expf(m - m_new);  // m==-inf -> exp(-inf)=0 ok
      if (m <= -3.0e38f) alpha = 0.f;
      m = m_new;
      const float p = valid ? __expf(sc - m_new) : 0.f;
      s = s * alpha + wave_reduce_sum(p);
#pragma unroll
      for (int j = 0; j < 8; ++j) o8[j] *= alpha;
      // ---- PV: PG positions per 16-B-per-lane iteration ----
      const uint4* vbase = reinterpret_cast<const uint4*>(
          vc + (((long long)page * KVH + kvh) * QSA_PAGE) * D);
      const int nvalid = min(seqlen - pi * QSA_PAGE, QSA_PAGE);
      const int niter = (nvalid + PG - 1) / PG;
#pragma unroll 4
      for (int it = 0; it < niter; ++it) {
        const int t = it * PG + pg;
        const float pt = __shfl(p, t, QSA_WAVE);  // p==0 beyond nvalid
        const uint4 v4 = vbase[(long long)t * DG + dg];
        unsigned int vv[4] = {v4.x, v4.y, v4.z, v4.w};
#pragma unroll
        for (int j = 0; j < 4; ++j) {
          float2 vf = bf16x2_to_f32x2(vv[j]);
          o8[2 * j] = fmaf(pt, vf.x, o8[2 * j]);
          o8[2 * j + 1] = fmaf(pt, vf.y, o8[2 * j + 1]);
        }
      }
    }
    // butterfly-reduce the PG position-group partials (same dg lanes)
#pragma unroll
    for (int off = DG; off < QSA_WAVE; off <<= 1) {
#pragma unroll
      for (int j = 0; j < 8; ++j) o8[j] += __shfl_xor(o8[j], off, QSA_WAVE);
    }
    const float inv = (s > 0.f) ? 1.f / s : 0.f;
    if (pg == 0) {
      uint4* orow = reinterpret_cast<uint4*>(out + ((long long)b * QH + qh) * D);
      uint4 packed;
      packed.x = f32x2_to_bf16x2(o8[0] * inv, o8[1] * inv);
      packed.y = f32x2_to_bf16x2(o8[2] * inv, o8[3] * inv);
      packed.z = f32x2_to_bf16x2(o8[4] * inv, o8[5] * inv);
      packed.w = f32x2_to_bf16x2(o8[6] * inv, o8[7] * inv);
      orow[dg] = packed;
    }
  }
}

// ---------------------------------------------------------------------------
// Flash-decoding context split: B*KVH workgroups underfill the 256-CU chip
// (e.g. B=24, KVH=8 -> 192 WGs, 4 waves each = terrible occupancy and the
// kernel runs latency-bound at ~8% of the HBM roofline).  Split the context
// into NS chunks -> grid B*KVH*NS (>=512 WGs, 2-4 blocks/CU), each workgroup
// produces an UNNORMALIZED partial (o, m, l) over its page range, and a tiny
// reduce kernel merges the NS partials per (b, head) row:
//   M = max m_s;  O = sum_s exp(m_s-M) o_s;  L = sum_s exp(m_s-M) l_s;
//   out = O / L.
// ---------------------------------------------------------------------------
template <int D>
__global__ void __launch_bounds__(256)
qsa_paged_attn_decode_split(const unsigned short* __restrict__ q,
                            const unsigned short* __restrict__ kc,
                            const unsigned short* __restrict__ vc,
                            const int* __restrict__ block_table,
                            const int* __restrict__ seq_lens,
                            float* __restrict__ part_o,    // [B, QH, NS, D]
                            float* __restrict__ part_ml,   // [B, QH, NS, 2]
                            float scale, int B, int QH, int KVH, int max_pages,
                            long long qstride, int NS) {
  const int split = blockIdx.x % NS;
  const int bk = blockIdx.x / NS;
  const int b = bk / KVH;
  const int kvh = bk % KVH;
  const int wave = threadIdx.x / QSA_WAVE;
  const int lane = threadIdx.x % QSA_WAVE;
  const int R = QH / KVH;
  const int seqlen = seq_lens[b];
  const int npages = (seqlen + QSA_PAGE - 1) / QSA_PAGE;
  const int chunk = (npages + NS - 1) / NS;
  const int p0 = split * chunk;
  const int p1 = min(npages, p0 + chunk);
  const int* btab = block_table + (long long)b * max_pages;

  for (int r = wave; r < R; r += 4) {
    const int qh = kvh * R + r;
    float* pml = part_ml + (((long long)b * QH + qh) * NS + split) * 2;
    float* po = part_o + (((long long)b * QH + qh) * NS + split) * D;
    if (seqlen <= 0 || p0 >= npages) {
      if (lane == 0) { pml[0] = -3.0e38f; pml[1] = 0.f; }
      continue;
    }
    unsigned int qpk[D / 2];
    const unsigned int* qsrc = reinterpret_cast<const unsigned int*>(
        q + (long long)b * qstride + (long long)qh * D);
#pragma unroll
    for (int i = 0; i < D / 2; ++i) qpk[i] = qsrc[i];

    constexpr int DG = D / 8;
    constexpr int PG = QSA_WAVE / DG;
    const int dg = lane % DG;
    const int pg = lane / DG;

    float m = -3.0e38f, s = 0.f;
    float o8[8];
#pragma unroll
    for (int j = 0; j < 8; ++j) o8[j] = 0.f;

    for (int pi = p0; pi < p1; ++pi) {
      const int page = btab[pi];
      const int pos = pi * QSA_PAGE + lane;
      const bool valid = pos < seqlen;
      float sc = 0.f;
      const uint4* kbase = reinterpret_cast<const uint4*>(
          kc + ((((long long)page * KVH + kvh) * (D / 8)) * QSA_PAGE) * 8);
#pragma unroll
      for (int d0 = 0; d0 < D / 8; ++d0) {
        uint4 kv4 = kbase[d0 * QSA_PAGE + lane];
        unsigned int kk[4] = {kv4.x, kv4.y, kv4.z, kv4.w};
#pragma unroll
        for (int j = 0; j < 4; ++j)   // v_dot2_f32_bf16: 2 MACs/instr
          sc = __builtin_amdgcn_fdot2_f32_bf16(
              as_bf16x2(kk[j]), as_bf16x2(qpk[d0 * 4 + j]), sc, false);
      }
      sc = valid ? sc * scale : -3.0e38f;
      const float pmax = wave_reduce_max(sc);
      const float m_new = fmaxf(m, pmax);
      float alpha = __expf(m - m_new);
      if (m <= -3.0e38f) alpha = 0.f;
      m = m_new;
      const float p = valid ? __expf(sc - m_new) : 0.f;
      s = s * alpha + wave_reduce_sum(p);
#pragma unroll
      for (int j = 0; j < 8; ++j) o8[j] *= alpha;
      const uint4* vbase = reinterpret_cast<const uint4*>(
          vc + (((long long)page * KVH + kvh) * QSA_PAGE) * D);
      const int nvalid = min(seqlen - pi * QSA_PAGE, QSA_PAGE);
      const int niter = (nvalid + PG - 1) / PG;
#pragma unroll 4
      for (int it = 0; it < niter; ++it) {
        const int t = it * PG + pg;
        const float pt = __shfl(p, t, QSA_WAVE);
        const uint4 v4 = vbase[(long long)t * DG + dg];
        unsigned int vv[4] = {v4.x, v4.y, v4.z, v4.w};
#pragma unroll
        for (int j = 0; j < 4; ++j) {
          float2 vf = bf16x2_to_f32x2(vv[j]);
          o8[2 * j] = fmaf(pt, vf.x, o8[2 * j]);
          o8[2 * j + 1] = fmaf(pt, vf.y, o8[2 * j + 1]);
        }
      }
    }
#pragma unroll
    for (int off = DG; off < QSA_WAVE; off <<= 1) {
#pragma unroll
      for (int j = 0; j < 8; ++j) o8[j] += __shfl_xor(o8[j], off, QSA_WAVE);
    }
    if (pg == 0) {
      // unnormalized partial in f32 (reduce kernel applies 1/L)
      float4* prow = reinterpret_cast<float4*>(po);
      prow[dg * 2] = make_float4(o8[0], o8[1], o8[2], o8[3]);
      prow[dg * 2 + 1] = make_float4(o8[4], o8[5], o8[6], o8[7]);
    }
    if (lane == 0) { pml[0] = m; pml[1] = s; }
  }
}

template <int D>
__global__ void __launch_bounds__(128)
qsa_attn_reduce(const float* __restrict__ part_o,   // [rows, NS, D]
                const float* __restrict__ part_ml,  // [rows, NS, 2]
                unsigned short* __restrict__ out,   // [rows, D]
                int NS) {
  const long long row = blockIdx.x;
  const int d = threadIdx.x;  // D threads
  const float* ml = part_ml + row * NS * 2;
  float M = -3.0e38f;
  for (int s = 0; s < NS; ++s)
    if (ml[2 * s + 1] > 0.f) M = fmaxf(M, ml[2 * s]);
  float L = 0.f, acc = 0.f;
  for (int s = 0; s < NS; ++s) {
    const float ls = ml[2 * s + 1];
    if (ls <= 0.f) continue;
    const float w = __expf(ml[2 * s] - M);
    L += w * ls;
    acc += w * part_o[(row * NS + s) * D + d];
  }
  const float v = (L > 0.f) ? acc / L : 0.f;
  out[row * D + d] = f32_to_bf16(v);
}

extern "C" void qsa_paged_attn_decode_launch(
    const unsigned short* q, const unsigned short* kc,
    const unsigned short* vc, const int* block_table, const int* seq_lens,
    unsigned short* out, float scale, int B, int QH, int KVH, int max_pages,
    int D, long long qstride, hipStream_t stream) {
  dim3 grid(B * KVH);
  dim3 block(256);
  if (D == 128) {
    hipLaunchKernelGGL((qsa_paged_attn_decode<128>), grid, block, 0, stream,
                       q, kc, vc, block_table, seq_lens, out, scale, B, QH,
                       KVH, max_pages, qstride);
  } else if (D == 64) {
    hipLaunchKernelGGL((qsa_paged_attn_decode<64>), grid, block, 0, stream,
                       q, kc, vc, block_table, seq_lens, out, scale, B, QH,
                       KVH, max_pages, qstride);
  }
}

extern "C" void qsa_paged_attn_decode_split_launch(
    const unsigned short* q, const unsigned short* kc,
    const unsigned short* vc, const int* block_table, const int* seq_lens,
    float* part_o, float* part_ml, unsigned short* out, float scale, int B,
    int QH, int KVH, int max_pages, int D, long long qstride, int NS,
    hipStream_t stream) {
  dim3 grid(B * KVH * NS);
  dim3 block(256);
  if (D == 128) {
    hipLaunchKernelGGL((qsa_paged_attn_decode_split<128>), grid, block, 0,
                       stream, q, kc, vc, block_table, seq_lens, part_o,
                       part_ml, scale, B, QH, KVH, max_pages, qstride, NS);
    hipLaunchKernelGGL((qsa_attn_reduce<128>), dim3(B * QH), dim3(128), 0,
                       stream, part_o, part_ml, out, NS);
  } else if (D == 64) {
    hipLaunchKernelGGL((qsa_paged_attn_decode_split<64>), grid, block, 0,
                       stream, q, kc, vc, block_table, seq_lens, part_o,
                       part_ml, scale, B, QH, KVH, max_pages, qstride, NS);
    hipLaunchKernelGGL((qsa_attn_reduce<64>), dim3(B * QH), dim3(64), 0,
                       stream, part_o, part_ml, out, NS);
  }
}

// ---------------------------------------------------------------------------
// KV-cache append: scatter the step's new k/v [B, KVH, D] into the paged
// cache at position seq_lens[b]-1 (called after RoPE, before attention).
// One block per (b, kvh); D threads.
// ---------------------------------------------------------------------------
__global__ void
qsa_kv_append(const unsigned short* __restrict__ knew,  // [B, KVH, D] (row stride kvstride)
              const unsigned short* __restrict__ vnew,
              unsigned short* __restrict__ kc, unsigned short* __restrict__ vc,
              const int* __restrict__ block_table, const int* __restrict__ seq_lens,
              int B, int KVH, int D, int max_pages, long long kvstride) {
  const int b = blockIdx.x / KVH;
  const int kvh = blockIdx.x % KVH;
  const int d = threadIdx.x;
  if (d >= D) return;
  const int pos = seq_lens[b] - 1;
  if (pos < 0) return;
  const int page = block_table[(long long)b * max_pages + pos / QSA_PAGE];
  const int pin = pos % QSA_PAGE;
  const unsigned short kv = knew[(long long)b * kvstride + (long long)kvh * D + d];
  const unsigned short vv = vnew[(long long)b * kvstride + (long long)kvh * D + d];
  // K layout [page, kvh, D/8, 64, 8]
  kc[((((long long)page * KVH + kvh) * (D / 8) + d / 8) * QSA_PAGE + pin) * 8 +
     d % 8] = kv;
  // V layout [page, kvh, 64, D]
  vc[(((long long)page * KVH + kvh) * QSA_PAGE + pin) * D + d] = vv;
}

// Prefill bulk variant: scatter T tokens' k/v [T, KVH, D] given their
// (seq position) mapping to pages via per-token slot ids precomputed on host:
// slot[t] = page * 64 + offset.
__global__ void
qsa_kv_scatter(const unsigned short* __restrict__ knew,  // [T, KVH, D]
               const unsigned short* __restrict__ vnew,
               unsigned short* __restrict__ kc, unsigned short* __restrict__ vc,
               const int* __restrict__ slots,  // [T]
               int T, int KVH, int D) {
  const long long t = blockIdx.x / KVH;
  const int kvh = blockIdx.x % KVH;
  const int d = threadIdx.x;
  if (t >= T || d >= D) return;
  const int slot = slots[t];
  const long long page = slot / QSA_PAGE;
  const int pin = slot % QSA_PAGE;
  const unsigned short kv = knew[(t * KVH + kvh) * D + d];
  const unsigned short vv = vnew[(t * KVH + kvh) * D + d];
  kc[(((page * KVH + kvh) * (D / 8) + d / 8) * QSA_PAGE + pin) * 8 + d % 8] = kv;
  vc[((page * KVH + kvh) * QSA_PAGE + pin) * D + d] = vv;
}

// Fused decode-step RoPE + KV append: rotates q in place, rotates k and
// writes it (and v) STRAIGHT into the paged cache — one kernel instead of
// rope + kv_append, and k never round-trips through HBM in its pre-cache
// form.  One block per (b, head) over QH q-heads + KVH kv-heads.
__global__ void
qsa_rope_kv_append(unsigned short* __restrict__ q,        // [B, QH, D]
                   const unsigned short* __restrict__ k,  // [B, KVH, D]
                   const unsigned short* __restrict__ v,  // [B, KVH, D]
                   unsigned short* __restrict__ kc, unsigned short* __restrict__ vc,
                   const float* __restrict__ cos_t, const float* __restrict__ sin_t,
                   const int* __restrict__ block_table,
                   const int* __restrict__ seq_lens,
                   int B, int QH, int KVH, int D, int max_pages,
                   long long qstride, long long kvstride) {
  const int nheads = QH + KVH;
  const int b = blockIdx.x / nheads;
  const int h = blockIdx.x % nheads;
  const int half = D / 2;
  const int d = threadIdx.x;
  const int pos = seq_lens[b] - 1;
  if (pos < 0) return;
  const long long toff = (long long)pos * half + (d % half);
  const float c = cos_t[toff], s = sin_t[toff];
  if (h < QH) {
    if (d >= half) return;
    unsigned short* base = q + (long long)b * qstride + (long long)h * D;
    const float x0 = bf16_to_f32(base[d]);
    const float x1 = bf16_to_f32(base[d + half]);
    base[d] = f32_to_bf16(x0 * c - x1 * s);
    base[d + half] = f32_to_bf16(x0 * s + x1 * c);
    return;
  }
  if (d >= D) return;
  const int kvh = h - QH;
  const int page = block_table[(long long)b * max_pages + pos / QSA_PAGE];
  const int pin = pos % QSA_PAGE;
  // k: rope pair (d, d+half) computed per thread d < half; threads
  // [half, D) carry the rotated upper half (recomputed: cheap VALU)
  const unsigned short* kbase =
      k + (long long)b * kvstride + (long long)kvh * D;
  const int dl = d % half;
  const float x0 = bf16_to_f32(kbase[dl]);
  const float x1 = bf16_to_f32(kbase[dl + half]);
  const float kr = (d < half) ? (x0 * c - x1 * s) : (x0 * s + x1 * c);
  kc[((((long long)page * KVH + kvh) * (D / 8) + d / 8) * QSA_PAGE + pin) * 8 +
     d % 8] = f32_to_bf16(kr);
  vc[(((long long)page * KVH + kvh) * QSA_PAGE + pin) * D + d] =
      v[(long long)b * kvstride + (long long)kvh * D + d];
}

extern "C" void qsa_rope_kv_append_launch(
    unsigned short* q, const unsigned short* k, const unsigned short* v,
    unsigned short* kc, unsigned short* vc, const float* cos_t,
    const float* sin_t, const int* block_table, const int* seq_lens, int B,
    int QH, int KVH, int D, int max_pages, long long qstride,
    long long kvstride, hipStream_t stream) {
  hipLaunchKernelGGL(qsa_rope_kv_append, dim3(B * (QH + KVH)), dim3(D), 0,
                     stream, q, k, v, kc, vc, cos_t, sin_t, block_table,
                     seq_lens, B, QH, KVH, D, max_pages, qstride, kvstride);
}

extern "C" void qsa_kv_append_launch(const unsigned short* knew,
                                     const unsigned short* vnew,
                                     unsigned short* kc, unsigned short* vc,
                                     const int* block_table,
                                     const int* seq_lens, int B, int KVH,
                                     int D, int max_pages, long long kvstride,
                                     hipStream_t stream) {
  hipLaunchKernelGGL(qsa_kv_append, dim3(B * KVH), dim3(D), 0, stream, knew,
                     vnew, kc, vc, block_table, seq_lens, B, KVH, D, max_pages,
                     kvstride);
}

extern "C" void qsa_kv_scatter_launch(const unsigned short* knew,
                                      const unsigned short* vnew,
                                      unsigned short* kc, unsigned short* vc,
                                      const int* slots, int T, int KVH, int D,
                                      hipStream_t stream) {
  hipLaunchKernelGGL(qsa_kv_scatter, dim3((long long)T * KVH), dim3(D), 0,
                     stream, knew, vnew, kc, vc, slots, T, KVH, D);
}
