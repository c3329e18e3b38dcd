// Paged-attention DECODE for CDNA4 (gfx950) — the K4 hot op, on MFMA.
//
// Replaces the reference's managed-LLM call (ML_PREDICT 'llm_textgen_model',
// terraform/core/main.tf:461) with on-GPU batched decode attention over a
// paged KV cache resident in 288 GB HBM3E.
//
// Design (flash-decoding, one WAVE per (batch b, kv head, context split)):
//   - All R GQA query heads of a kv head compute TOGETHER on the matrix
//     cores: scores via mfma_f32_16x16x32_bf16 with A = K-tile
//     [16 pos x 32 k] and B = Q [32 k x 16 heads] (heads pad to 16), PV
//     via A = P [16 heads x 32 pos] (packed bf16, redistributed with 8
//     cross-lane shuffles) and B = V^T [32 pos x 16 dims].  A VALU
//     decode kernel pays ~16 VALU per 16 B of KV; this pays ~1 MFMA per
//     1 KiB plus the online-softmax tail.
//   - K cache layout [page, kvh, D/8, 64, 8] (d-major x8): the A-fragment
//     read (lane = pos x k-slice) is one fully coalesced 1 KiB wave load.
//   - V cache layout [page, kvh, D, 64] (TRANSPOSED, pos minor): the PV
//     B-fragment read (lane = dim x pos-slice) is also one coalesced 1 KiB
//     wave load.  Every KV byte is read exactly ONCE per (b, kvh) — the
//     per-head VALU design re-read it R times.
//   - Context splits fill the chip (grid = B*KVH*NS >= ~2k waves);
//     unnormalized partials (o, m, l) merge in qsa_attn_reduce.
//
// The online softmax runs per head-column in registers: column max/sum
// reduce with two shfl_xor hops (the 4 hi-lane groups of a column).
#include "common.h"

#define QSA_PAGE 64

using bf16x8_a = __attribute__((ext_vector_type(8))) short;
using f32x4_a = __attribute__((ext_vector_type(4))) float;

template <int D>
__global__ void __launch_bounds__(256)
qsa_paged_attn_mfma(const unsigned short* __restrict__ q,   // [B, QH, D]
                    const unsigned short* __restrict__ kc,  // [P, KVH, D/8, 64, 8]
                    const unsigned short* __restrict__ vc,  // [P, KVH, D, 64]
                    const int* __restrict__ block_table,    // [B, max_pages]
                    const int* __restrict__ seq_lens,       // [B]
                    float* __restrict__ part_o,              // [B, QH, NS, D]
                    float* __restrict__ part_ml,             // [B, QH, NS, 2]
                    float scale, int B, int QH, int KVH, int max_pages,
                    long long qstride, int NS) {
  constexpr int KSTEPS = D / 32;      // QK^T k-steps (4 for D=128)
  constexpr int DTILES = D / 16;      // PV d-tiles (8 for D=128)
  const int wave = threadIdx.x >> 6;
  const int lane = threadIdx.x & 63;
  const int nsb = (NS + 3) / 4;       // split-blocks per (b, kvh)
  const int bk = blockIdx.x / nsb;
  const int split = (blockIdx.x % nsb) * 4 + wave;
  const int b = bk / KVH;
  const int kvh = bk % KVH;
  const int R = QH / KVH;
  if (split >= NS) return;

  const int seqlen = seq_lens[b];
  const int npages = (seqlen + QSA_PAGE - 1) / QSA_PAGE;
  const int chunk = (npages + NS - 1) / NS;
  const int p0 = split * chunk;
  const int p1 = min(npages, p0 + chunk);
  const int col = lane & 15;          // head col (scores) / dim col (PV)
  const int hi = lane >> 4;           // 4 hi-lane groups

  // every (head) slot writes its partial, real or empty
  float* pml_base = part_ml + (((long long)b * QH + kvh * R) * NS + split) * 2;
  if (seqlen <= 0 || p0 >= npages) {
    if (lane < R)
      *reinterpret_cast<float2*>(pml_base + (long long)lane * NS * 2) =
          make_float2(-3.0e38f, 0.f);
    return;
  }

  // ---- Q B-fragments: lane holds Q[head=col][k = ks*32 + hi*8 + e] ----
  const int qh_l = kvh * R + min(col, R - 1);   // clamp: pad heads compute
  const unsigned short* qrow = q + (long long)b * qstride + (long long)qh_l * D;
  bf16x8_a qf[KSTEPS];
#pragma unroll
  for (int ks = 0; ks < KSTEPS; ++ks)
    qf[ks] = *reinterpret_cast<const bf16x8_a*>(qrow + ks * 32 + hi * 8);

  float m = -3.0e38f, lsum = 0.f;
  f32x4_a o_acc[DTILES];
#pragma unroll
  for (int dt = 0; dt < DTILES; ++dt)
    o_acc[dt] = (f32x4_a){0.f, 0.f, 0.f, 0.f};

  const int* btab = block_table + (long long)b * max_pages;
  for (int pi = p0; pi < p1; ++pi) {
    const int page = btab[pi];
    const unsigned short* kbase =
        kc + ((((long long)page * KVH + kvh) * (D / 8)) * QSA_PAGE) * 8;
    const unsigned short* vbase =
        vc + (((long long)page * KVH + kvh) * D) * QSA_PAGE;
    const int nvalid = min(seqlen - pi * QSA_PAGE, QSA_PAGE);

    // ---- scores: 4 pos-tiles x KSTEPS MFMAs ---------------------------
    // A-frag: lane holds K[pos = pt*16 + col][k = ks*32 + hi*8 + e];
    // K layout element (pos, k) at ((k/8)*64 + pos)*8 + k%8 ->
    // 16 B at ((ks*4 + hi)*64 + pos)*8.
    f32x4_a sc[4];
#pragma unroll
    for (int pt = 0; pt < 4; ++pt) {
      f32x4_a acc = {0.f, 0.f, 0.f, 0.f};
      const int pos = pt * 16 + col;
#pragma unroll
      for (int ks = 0; ks < KSTEPS; ++ks) {
        const bf16x8_a kf = *reinterpret_cast<const bf16x8_a*>(
            kbase + (((long long)(ks * 4 + hi) * QSA_PAGE) + pos) * 8);
        acc = __builtin_amdgcn_mfma_f32_16x16x32_bf16(kf, qf[ks], acc,
                                                      0, 0, 0);
      }
      sc[pt] = acc;
    }
    // scale + mask invalid positions; C layout: row pos = hi*4 + r,
    // col head = lane&15 -> lane's r-th value is position pt*16 + hi*4 + r
    float pagemax = -3.0e38f;
#pragma unroll
    for (int pt = 0; pt < 4; ++pt) {
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const int pos = pt * 16 + hi * 4 + r;
        float v = (pos < nvalid) ? sc[pt][r] * scale : -3.0e38f;
        sc[pt][r] = v;
        pagemax = fmaxf(pagemax, v);
      }
    }
    // column (per-head) max across the 4 hi groups
    pagemax = fmaxf(pagemax, __shfl_xor(pagemax, 16, QSA_WAVE));
    pagemax = fmaxf(pagemax, __shfl_xor(pagemax, 32, QSA_WAVE));
    const float m_new = fmaxf(m, pagemax);
    float alpha = __expf(m - m_new);
    if (m <= -3.0e38f) alpha = 0.f;
    m = m_new;
    // o_acc element (dt, r) belongs to OUTPUT ROW hi*4 + r (the PV
    // C-layout), while alpha lives per score COLUMN (this lane's l&15):
    // fetch each row's own alpha from the lane holding that column —
    // using the lane's own alpha zeroes/mis-scales other rows' history
    // whenever per-row maxima diverge across pages
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      const float alpha_r = __shfl(alpha, hi * 4 + r, 16);
#pragma unroll
      for (int dt = 0; dt < DTILES; ++dt) o_acc[dt][r] *= alpha_r;
    }
    // P = exp(sc - m) (invalid -> 0), packed bf16x2 per tile quad
    unsigned int ppk[8];   // tile pt -> 2 uints (4 bf16 = quads r0..3)
    float psum = 0.f;
#pragma unroll
    for (int pt = 0; pt < 4; ++pt) {
      float p0f = 0.f, p1f = 0.f, p2f = 0.f, p3f = 0.f;
      if (sc[pt][0] > -1.0e38f) p0f = __expf(sc[pt][0] - m_new);
      if (sc[pt][1] > -1.0e38f) p1f = __expf(sc[pt][1] - m_new);
      if (sc[pt][2] > -1.0e38f) p2f = __expf(sc[pt][2] - m_new);
      if (sc[pt][3] > -1.0e38f) p3f = __expf(sc[pt][3] - m_new);
      psum += p0f + p1f + p2f + p3f;
      ppk[pt * 2] = f32x2_to_bf16x2(p0f, p1f);
      ppk[pt * 2 + 1] = f32x2_to_bf16x2(p2f, p3f);
    }
    psum += __shfl_xor(psum, 16, QSA_WAVE);
    psum += __shfl_xor(psum, 32, QSA_WAVE);
    lsum = lsum * alpha + psum;

    // ---- PV: two 32-pos halves ---------------------------------------
    // A-frag needs lane (head=col, hi) to hold P[head][pos = hi*8 + e]:
    // packed pairs come from source lanes (same col) with hi' = (2*hi)&3
    // and (2*hi+1)&3, tile = half*2 + (hi>>1).
#pragma unroll
    for (int half = 0; half < 2; ++half) {
      // lane (col, hi) needs P[head=col][pos = half*32 + hi*8 + e]:
      // tile T = half*2 + (hi>>1), quads (hi&1)*2 and (hi&1)*2+1 — i.e.
      // source lanes (col, g) with g = (2*hi)&3 and (2*hi+1)&3.  __shfl
      // evaluates the value expression in EACH lane, so the register
      // index must be compile-time: shuffle both candidate tiles'
      // packed regs and select by hi (guide rule 20: runtime-indexed
      // arrays spill to scratch; and a runtime index would read the
      // SOURCE lane's differently-computed tile).
      const int src_a = col + (((2 * hi) & 3) << 4);
      const int src_b = col + (((2 * hi + 1) & 3) << 4);
      const int tA = half * 2;           // tile for hi 0,1
      const int tB = half * 2 + 1;       // tile for hi 2,3
      const unsigned int a0A = __shfl(ppk[tA * 2], src_a, QSA_WAVE);
      const unsigned int a1A = __shfl(ppk[tA * 2 + 1], src_a, QSA_WAVE);
      const unsigned int b0A = __shfl(ppk[tA * 2], src_b, QSA_WAVE);
      const unsigned int b1A = __shfl(ppk[tA * 2 + 1], src_b, QSA_WAVE);
      const unsigned int a0B = __shfl(ppk[tB * 2], src_a, QSA_WAVE);
      const unsigned int a1B = __shfl(ppk[tB * 2 + 1], src_a, QSA_WAVE);
      const unsigned int b0B = __shfl(ppk[tB * 2], src_b, QSA_WAVE);
      const unsigned int b1B = __shfl(ppk[tB * 2 + 1], src_b, QSA_WAVE);
      const bool lo = hi < 2;
      bf16x8_a pa;
      unsigned int* pa_u = reinterpret_cast<unsigned int*>(&pa);
      pa_u[0] = lo ? a0A : a0B;
      pa_u[1] = lo ? a1A : a1B;
      pa_u[2] = lo ? b0A : b0B;
      pa_u[3] = lo ? b1A : b1B;
#pragma unroll
      for (int dt = 0; dt < DTILES; ++dt) {
        // B-frag: lane holds V^T[pos = half*32 + hi*8 + e][d = dt*16+col]
        // = vc[.., d, pos]: 16 B at (d*64 + half*32 + hi*8)
        const bf16x8_a vf = *reinterpret_cast<const bf16x8_a*>(
            vbase + ((long long)(dt * 16 + col) * QSA_PAGE) + half * 32 +
            hi * 8);
        o_acc[dt] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(pa, vf,
                                                            o_acc[dt],
                                                            0, 0, 0);
      }
    }
  }

  // ---- store partials: O C-layout row = head = hi*4 + r, col = dim ----
  // lane (hi, col): holds O[head hi*4+r][d = dt*16 + col]
#pragma unroll
  for (int r = 0; r < 4; ++r) {
    const int head = hi * 4 + r;
    if (head < R) {
      float* po = part_o +
          (((long long)b * QH + kvh * R + head) * NS + split) * D;
#pragma unroll
      for (int dt = 0; dt < DTILES; ++dt)
        po[dt * 16 + col] = o_acc[dt][r];
    }
  }
  if (lane < R)
    *reinterpret_cast<float2*>(pml_base + (long long)lane * NS * 2) =
        make_float2(m, lsum);
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

extern "C" void qsa_paged_attn_mfma_launch(
    const unsigned short* q, const unsigned short* kc,
    const unsigned short* vc, const int* block_table, const int* seq_lens,
    float* part_o, float* part_ml, unsigned short* out, float scale, int B,
    int QH, int KVH, int max_pages, int D, long long qstride, int NS,
    hipStream_t stream) {
  const int nsb = (NS + 3) / 4;
  dim3 grid(B * KVH * nsb);
  dim3 block(256);
  if (D == 128) {
    hipLaunchKernelGGL((qsa_paged_attn_mfma<128>), grid, block, 0, stream,
                       q, kc, vc, block_table, seq_lens, part_o, part_ml,
                       scale, B, QH, KVH, max_pages, qstride, NS);
    hipLaunchKernelGGL((qsa_attn_reduce<128>), dim3(B * QH), dim3(128), 0,
                       stream, part_o, part_ml, out, NS);
  } else if (D == 64) {
    hipLaunchKernelGGL((qsa_paged_attn_mfma<64>), grid, block, 0, stream,
                       q, kc, vc, block_table, seq_lens, part_o, part_ml,
                       scale, B, QH, KVH, max_pages, qstride, NS);
    hipLaunchKernelGGL((qsa_attn_reduce<64>), dim3(B * QH), dim3(64), 0,
                       stream, part_o, part_ml, out, NS);
  }
}

// ---------------------------------------------------------------------------
// KV-cache writers.  K layout [page, kvh, D/8, 64, 8]; V layout
// [page, kvh, D, 64] (transposed for the PV B-fragment stream).
// ---------------------------------------------------------------------------

// Fused decode-step RoPE + KV append: rotates q in place, rotates k and
// writes it (and v) STRAIGHT into the paged cache.
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
  const unsigned short* kbase =
      k + (long long)b * kvstride + (long long)kvh * D;
  const int dl = d % half;
  const float x0 = bf16_to_f32(kbase[dl]);
  const float x1 = bf16_to_f32(kbase[dl + half]);
  const float kr = (d < half) ? (x0 * c - x1 * s) : (x0 * s + x1 * c);
  kc[((((long long)page * KVH + kvh) * (D / 8) + d / 8) * QSA_PAGE + pin) * 8 +
     d % 8] = f32_to_bf16(kr);
  vc[(((long long)page * KVH + kvh) * D + d) * QSA_PAGE + pin] =
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

// Un-fused single-step append (numerics tests / CPU-parity surface).
__global__ void
qsa_kv_append(const unsigned short* __restrict__ knew,  // [B, KVH, D]
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
  kc[((((long long)page * KVH + kvh) * (D / 8) + d / 8) * QSA_PAGE + pin) * 8 +
     d % 8] = kv;
  vc[(((long long)page * KVH + kvh) * D + d) * QSA_PAGE + pin] = vv;
}

// Prefill bulk scatter: T tokens' k/v by precomputed slot ids
// (slot = page*64 + offset).  The V^T write is a 2-byte-per-thread scatter;
// consecutive tokens of a page share cache lines per d, so L2 absorbs it.
__global__ void
qsa_kv_scatter(const unsigned short* __restrict__ knew,  // [T, KVH, D] (row stride kvstride)
               const unsigned short* __restrict__ vnew,
               unsigned short* __restrict__ kc, unsigned short* __restrict__ vc,
               const int* __restrict__ slots,  // [T]
               int T, int KVH, int D, long long kvstride) {
  const long long t = blockIdx.x / KVH;
  const int kvh = blockIdx.x % KVH;
  const int d = threadIdx.x;
  if (t >= T || d >= D) return;
  const int slot = slots[t];
  const long long page = slot / QSA_PAGE;
  const int pin = slot % QSA_PAGE;
  const unsigned short kv = knew[t * kvstride + (long long)kvh * D + d];
  const unsigned short vv = vnew[t * kvstride + (long long)kvh * D + d];
  kc[(((page * KVH + kvh) * (D / 8) + d / 8) * QSA_PAGE + pin) * 8 + d % 8] = kv;
  vc[((page * KVH + kvh) * D + d) * QSA_PAGE + pin] = vv;
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
                                      long long kvstride, hipStream_t stream) {
  hipLaunchKernelGGL(qsa_kv_scatter, dim3((long long)T * KVH), dim3(D), 0,
                     stream, knew, vnew, kc, vc, slots, T, KVH, D, kvstride);
}

// ---------------------------------------------------------------------------
// Varlen flash PREFILL attention over the paged cache (K4 prefill path).
//
// One WAVE per (16-row q-block, q head): streams the sequence's K/V pages
// directly from the paged cache (K [P,KVH,D/8,64,8], V^T [P,KVH,D,64] —
// the same coalesced MFMA fragment reads as the decode kernel), online
// softmax per q-row column, causal masking against the row's absolute
// position (cached prefix start + row).  No K/V gather, no padding, no
// materialized score matrix: each KV byte is read once per wave and the
// bmm+softmax+bmm chain collapses into one kernel per layer.
//
// Host passes per-q-block maps (item id, first q position) plus per-item
// row offset / cached start / new length and the padded page table.
//
// CAUSAL=false is the ENCODER path (K1, models/encoder.py): bidirectional
// attention over the full item — every row attends to every valid key —
// with the same varlen layout and paged K/V fragment reads.
// ---------------------------------------------------------------------------
template <int D, bool CAUSAL>
__global__ void __launch_bounds__(256)
qsa_paged_attn_prefill(const unsigned short* __restrict__ q,   // [T, QH, D] strided
                       const unsigned short* __restrict__ kc,
                       const unsigned short* __restrict__ vc,
                       const int* __restrict__ block_table,    // [nb, npmax]
                       const int* __restrict__ qb_item,        // [QB]
                       const int* __restrict__ qb_pos0,        // [QB]
                       const int* __restrict__ item_off,       // [nb] row offset
                       const int* __restrict__ item_start,     // [nb] cached
                       const int* __restrict__ item_len,       // [nb] new rows
                       unsigned short* __restrict__ out,       // [T, QH*D]
                       float scale, int QB, int QH, int KVH, int npmax,
                       long long qstride) {
  constexpr int KSTEPS = D / 32;
  constexpr int DTILES = D / 16;
  const int lane = threadIdx.x & 63;
  const int gw = blockIdx.x * 4 + (threadIdx.x >> 6);  // global wave id
  // the 4 waves of a workgroup are 4 consecutive heads of the SAME
  // q-block (and, at R>=4, the same kv head): they stream identical
  // K/V pages, so keeping them in lockstep (per-page barrier below)
  // turns 3 of the 4 streams into L1 hits
  const bool live = gw < QB * QH;
  const int gws = live ? gw : QB * QH - 1;   // dead waves shadow the last
  const int qb = gws / QH;
  const int qh = gws % QH;
  const int R = QH / KVH;
  const int kvh = qh / R;
  const int item = qb_item[qb];
  const int pos0 = qb_pos0[qb];          // first new-row index in the item
  const int n_i = item_len[item];
  const int start = item_start[item];
  const int off = item_off[item];
  const int col = lane & 15;             // q-row column in the score tiles
  const int hi = lane >> 4;

  // ---- Q B-fragments: lane holds Q[row pos0+col][k = ks*32 + hi*8+e] --
  const int qrow_l = min(pos0 + col, n_i - 1);   // clamp pad rows
  const unsigned short* qrow =
      q + (long long)(off + qrow_l) * qstride + (long long)qh * D;
  bf16x8_a qf[KSTEPS];
#pragma unroll
  for (int ks = 0; ks < KSTEPS; ++ks)
    qf[ks] = *reinterpret_cast<const bf16x8_a*>(qrow + ks * 32 + hi * 8);

  float m = -3.0e38f, lsum = 0.f;
  f32x4_a o_acc[DTILES];
#pragma unroll
  for (int dt = 0; dt < DTILES; ++dt)
    o_acc[dt] = (f32x4_a){0.f, 0.f, 0.f, 0.f};

  // attention horizon: causal -> the block's LAST row's position;
  // bidirectional -> the whole item
  const int max_abs = CAUSAL ? start + min(pos0 + 15, n_i - 1)
                             : start + n_i - 1;
  const int npages = (max_abs + QSA_PAGE) / QSA_PAGE;  // ceil(max_abs+1/64)
  const int* btab = block_table + (long long)item * npmax;

  for (int pi = 0; pi < npages; ++pi) {
    __syncthreads();   // lockstep the 4 head-waves on the shared page
    const int page = btab[pi];
    const unsigned short* kbase =
        kc + ((((long long)page * KVH + kvh) * (D / 8)) * QSA_PAGE) * 8;
    const unsigned short* vbase =
        vc + (((long long)page * KVH + kvh) * D) * QSA_PAGE;
    const int nvalid = min(max_abs + 1 - pi * QSA_PAGE, QSA_PAGE);

    f32x4_a sc[4];
#pragma unroll
    for (int pt = 0; pt < 4; ++pt) {
      f32x4_a acc = {0.f, 0.f, 0.f, 0.f};
      const int pos = pt * 16 + col;
#pragma unroll
      for (int ks = 0; ks < KSTEPS; ++ks) {
        const bf16x8_a kf = *reinterpret_cast<const bf16x8_a*>(
            kbase + (((long long)(ks * 4 + hi) * QSA_PAGE) + pos) * 8);
        acc = __builtin_amdgcn_mfma_f32_16x16x32_bf16(kf, qf[ks], acc,
                                                      0, 0, 0);
      }
      sc[pt] = acc;
    }
    // mask: kpos (abs) must be <= start + qrow (abs) and < nvalid bound;
    // C layout row kpos = pt*16 + hi*4 + r, col = q-row
    const int qabs = start + pos0 + col;          // this column's row
    float pagemax = -3.0e38f;
#pragma unroll
    for (int pt = 0; pt < 4; ++pt) {
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const int kpos = pt * 16 + hi * 4 + r;
        const int kabs = pi * QSA_PAGE + kpos;
        const bool ok = (kpos < nvalid) && (!CAUSAL || kabs <= qabs) &&
                        (pos0 + col < n_i);
        float v = ok ? sc[pt][r] * scale : -3.0e38f;
        sc[pt][r] = v;
        pagemax = fmaxf(pagemax, v);
      }
    }
    pagemax = fmaxf(pagemax, __shfl_xor(pagemax, 16, QSA_WAVE));
    pagemax = fmaxf(pagemax, __shfl_xor(pagemax, 32, QSA_WAVE));
    // skip only when EVERY column is masked: the skip must be
    // wave-uniform — MFMA reads all 64 lanes' source registers
    // regardless of EXEC, so divergence around the PV MFMAs corrupts
    // active lanes' products
    if (__all(pagemax <= -3.0e38f)) continue;
    const float m_new = fmaxf(m, pagemax);
    float alpha = __expf(m - m_new);
    if (m <= -3.0e38f) alpha = 0.f;
    m = m_new;
    // o_acc element (dt, r) belongs to OUTPUT ROW hi*4 + r (the PV
    // C-layout), while alpha lives per score COLUMN (this lane's l&15):
    // fetch each row's own alpha from the lane holding that column —
    // using the lane's own alpha zeroes/mis-scales other rows' history
    // whenever per-row maxima diverge across pages
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      const float alpha_r = __shfl(alpha, hi * 4 + r, 16);
#pragma unroll
      for (int dt = 0; dt < DTILES; ++dt) o_acc[dt][r] *= alpha_r;
    }
    unsigned int ppk[8];
    float psum = 0.f;
#pragma unroll
    for (int pt = 0; pt < 4; ++pt) {
      float p0f = 0.f, p1f = 0.f, p2f = 0.f, p3f = 0.f;
      if (sc[pt][0] > -1.0e38f) p0f = __expf(sc[pt][0] - m_new);
      if (sc[pt][1] > -1.0e38f) p1f = __expf(sc[pt][1] - m_new);
      if (sc[pt][2] > -1.0e38f) p2f = __expf(sc[pt][2] - m_new);
      if (sc[pt][3] > -1.0e38f) p3f = __expf(sc[pt][3] - m_new);
      psum += p0f + p1f + p2f + p3f;
      ppk[pt * 2] = f32x2_to_bf16x2(p0f, p1f);
      ppk[pt * 2 + 1] = f32x2_to_bf16x2(p2f, p3f);
    }
    psum += __shfl_xor(psum, 16, QSA_WAVE);
    psum += __shfl_xor(psum, 32, QSA_WAVE);
    lsum = lsum * alpha + psum;

#pragma unroll
    for (int half = 0; half < 2; ++half) {
      const int src_a = col + (((2 * hi) & 3) << 4);
      const int src_b = col + (((2 * hi + 1) & 3) << 4);
      const int tA = half * 2;
      const int tB = half * 2 + 1;
      const unsigned int a0A = __shfl(ppk[tA * 2], src_a, QSA_WAVE);
      const unsigned int a1A = __shfl(ppk[tA * 2 + 1], src_a, QSA_WAVE);
      const unsigned int b0A = __shfl(ppk[tA * 2], src_b, QSA_WAVE);
      const unsigned int b1A = __shfl(ppk[tA * 2 + 1], src_b, QSA_WAVE);
      const unsigned int a0B = __shfl(ppk[tB * 2], src_a, QSA_WAVE);
      const unsigned int a1B = __shfl(ppk[tB * 2 + 1], src_a, QSA_WAVE);
      const unsigned int b0B = __shfl(ppk[tB * 2], src_b, QSA_WAVE);
      const unsigned int b1B = __shfl(ppk[tB * 2 + 1], src_b, QSA_WAVE);
      const bool lo = hi < 2;
      bf16x8_a pa;
      unsigned int* pa_u = reinterpret_cast<unsigned int*>(&pa);
      pa_u[0] = lo ? a0A : a0B;
      pa_u[1] = lo ? a1A : a1B;
      pa_u[2] = lo ? b0A : b0B;
      pa_u[3] = lo ? b1A : b1B;
#pragma unroll
      for (int dt = 0; dt < DTILES; ++dt) {
        const bf16x8_a vf = *reinterpret_cast<const bf16x8_a*>(
            vbase + ((long long)(dt * 16 + col) * QSA_PAGE) + half * 32 +
            hi * 8);
        o_acc[dt] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(pa, vf,
                                                            o_acc[dt],
                                                            0, 0, 0);
      }
    }
  }

  // ---- epilogue: O C-layout row = q-row = hi*4 + r, col = dim ----------
  // per-row 1/l lives in the lanes of its COLUMN; fetch via shfl
#pragma unroll
  for (int r = 0; r < 4; ++r) {
    const int qr = hi * 4 + r;               // q-row this lane holds
    const float lr = __shfl(lsum, qr, 16);   // lanes 0-15 hold cols 0-15
    const float inv = (lr > 0.f) ? 1.f / lr : 0.f;
    if (pos0 + qr < n_i) {
      unsigned short* orow =
          out + (long long)(off + pos0 + qr) * (QH * D) + (long long)qh * D;
#pragma unroll
      for (int dt = 0; dt < DTILES; ++dt)
        orow[dt * 16 + col] = f32_to_bf16(o_acc[dt][r] * inv);
    }
  }
}

extern "C" void qsa_paged_attn_prefill_launch(
    const unsigned short* q, const unsigned short* kc,
    const unsigned short* vc, const int* block_table, const int* qb_item,
    const int* qb_pos0, const int* item_off, const int* item_start,
    const int* item_len, unsigned short* out, float scale, int QB, int QH,
    int KVH, int npmax, int D, long long qstride, int causal,
    hipStream_t stream) {
  const long long waves = (long long)QB * QH;
  dim3 grid((unsigned)((waves + 3) / 4));
#define QSA_PREFILL_CASE(DD, CC)                                          \
  if (D == DD && causal == (int)CC) {                                     \
    hipLaunchKernelGGL((qsa_paged_attn_prefill<DD, CC>), grid, dim3(256), \
                       0, stream, q, kc, vc, block_table, qb_item,        \
                       qb_pos0, item_off, item_start, item_len, out,      \
                       scale, QB, QH, KVH, npmax, qstride);               \
    return;                                                               \
  }
  QSA_PREFILL_CASE(128, true) QSA_PREFILL_CASE(128, false)
  QSA_PREFILL_CASE(64, true) QSA_PREFILL_CASE(64, false)
#undef QSA_PREFILL_CASE
}
