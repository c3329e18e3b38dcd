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
qsa_paged_attn_decode(const unsigned short* __restrict__ q,   // [B, QH, D]
                      const unsigned short* __restrict__ kc,  // [P, KVH, D/8, 64, 8]
                      const unsigned short* __restrict__ vc,  // [P, KVH, 64, D]
                      const int* __restrict__ block_table,    // [B, max_pages]
                      const int* __restrict__ seq_lens,       // [B]
                      unsigned short* __restrict__ out,       // [B, QH, D]
                      float scale, int B, int QH, int KVH, int max_pages) {
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
        q + ((long long)b * QH + qh) * D);
#pragma unroll
    for (int i = 0; i < D / 2; ++i) qpk[i] = qsrc[i];

    float m = -3.0e38f, s = 0.f;
    float o0 = 0.f, o1 = 0.f;  // lane-owned dims 2*lane, 2*lane+1

    for (int pi = 0; pi < npages; ++pi) {
      const int page = btab[pi];
      const int pos = pi * QSA_PAGE + lane;
      const bool valid = pos < seqlen;
      // ---- score: dot(q, K[:, pos]) over d-slices of 8 ----
      float sc = 0.f;
      const uint4* kbase = reinterpret_cast<const uint4*>(
          kc + ((((long long)page * KVH + kvh) * (D / 8)) * QSA_PAGE) * 8);
      // slice d0 stride in uint4 units: QSA_PAGE (64 lanes * 16B)
#pragma unroll
      for (int d0 = 0; d0 < D / 8; ++d0) {
        uint4 kv4 = kbase[d0 * QSA_PAGE + lane];
        unsigned int kk[4] = {kv4.x, kv4.y, kv4.z, kv4.w};
#pragma unroll
        for (int j = 0; j < 4; ++j) {
          float2 kf = bf16x2_to_f32x2(kk[j]);
          float2 qf = bf16x2_to_f32x2(qpk[d0 * 4 + j]);
          sc = fmaf(kf.x, qf.x, sc);
          sc = fmaf(kf.y, qf.y, sc);
        }
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
      o0 *= alpha;
      o1 *= alpha;
      // ---- PV: lane owns dims (2l, 2l+1); prob via shfl broadcast ----
      const unsigned short* vbase =
          vc + (((long long)page * KVH + kvh) * QSA_PAGE) * D;
      const int nvalid = min(seqlen - pi * QSA_PAGE, QSA_PAGE);
#pragma unroll 4
      for (int t = 0; t < nvalid; ++t) {
        const float pt = __shfl(p, t, QSA_WAVE);
        if (lane * 2 < D) {
          const unsigned int vpk = *reinterpret_cast<const unsigned int*>(
              vbase + (long long)t * D + lane * 2);
          float2 vf = bf16x2_to_f32x2(vpk);
          o0 = fmaf(pt, vf.x, o0);
          o1 = fmaf(pt, vf.y, o1);
        }
      }
    }
    const float inv = (s > 0.f) ? 1.f / s : 0.f;
    unsigned int* orow = reinterpret_cast<unsigned int*>(
        out + ((long long)b * QH + qh) * D);
    if (lane * 2 < D) orow[lane] = f32x2_to_bf16x2(o0 * inv, o1 * inv);
  }
}

extern "C" void qsa_paged_attn_decode_launch(
    const unsigned short* q, const unsigned short* kc,
    const unsigned short* vc, const int* block_table, const int* seq_lens,
    unsigned short* out, float scale, int B, int QH, int KVH, int max_pages,
    int D, hipStream_t stream) {
  dim3 grid(B * KVH);
  dim3 block(256);
  if (D == 128) {
    hipLaunchKernelGGL((qsa_paged_attn_decode<128>), grid, block, 0, stream,
                       q, kc, vc, block_table, seq_lens, out, scale, B, QH,
                       KVH, max_pages);
  } else if (D == 64) {
    hipLaunchKernelGGL((qsa_paged_attn_decode<64>), grid, block, 0, stream,
                       q, kc, vc, block_table, seq_lens, out, scale, B, QH,
                       KVH, max_pages);
  }
}

// ---------------------------------------------------------------------------
// KV-cache append: scatter the step's new k/v [B, KVH, D] into the paged
// cache at position seq_lens[b]-1 (called after RoPE, before attention).
// One block per (b, kvh); D threads.
// ---------------------------------------------------------------------------
__global__ void
qsa_kv_append(const unsigned short* __restrict__ knew,  // [B, KVH, D]
              const unsigned short* __restrict__ vnew,  // [B, KVH, D]
              unsigned short* __restrict__ kc, unsigned short* __restrict__ vc,
              const int* __restrict__ block_table, const int* __restrict__ seq_lens,
              int B, int KVH, int D, int max_pages) {
  const int b = blockIdx.x / KVH;
  const int kvh = blockIdx.x % KVH;
  const int d = threadIdx.x;
  if (d >= D) return;
  const int pos = seq_lens[b] - 1;
  if (pos < 0) return;
  const int page = block_table[(long long)b * max_pages + pos / QSA_PAGE];
  const int pin = pos % QSA_PAGE;
  const unsigned short kv = knew[((long long)b * KVH + kvh) * D + d];
  const unsigned short vv = vnew[((long long)b * KVH + kvh) * D + d];
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

extern "C" void qsa_kv_append_launch(const unsigned short* knew,
                                     const unsigned short* vnew,
                                     unsigned short* kc, unsigned short* vc,
                                     const int* block_table,
                                     const int* seq_lens, int B, int KVH,
                                     int D, int max_pages, hipStream_t stream) {
  hipLaunchKernelGGL(qsa_kv_append, dim3(B * KVH), dim3(D), 0, stream, knew,
                     vnew, kc, vc, block_table, seq_lens, B, KVH, D, max_pages);
}

extern "C" void qsa_kv_scatter_launch(const unsigned short* knew,
                                      const unsigned short* vnew,
                                      unsigned short* kc, unsigned short* vc,
                                      const int* slots, int T, int KVH, int D,
                                      hipStream_t stream) {
  hipLaunchKernelGGL(qsa_kv_scatter, dim3((long long)T * KVH), dim3(D), 0,
                     stream, knew, vnew, kc, vc, slots, T, KVH, D);
}
