// HBM-resident vector search: exact cosine top-k (the K2 hot op).
//
// Replaces the reference's MongoDB Atlas $vectorSearch (lab2 main.tf:215:
// 1536-dim cosine, k=3, numCandidates=500) with an exact scan of the
// normalized index matrix in HBM.
//
// Stage 1: grid (Q, ceil(N / DOCS_PER_BLOCK)); each block stages the query
// in LDS, each of 4 waves computes one doc dot-product at a time (lanes
// split D, fully coalesced row reads, shfl reduce), maintains a per-wave
// top-k and emits a block top-k to the candidate buffer.
// Stage 2: one block per query reduces its candidate buffer to the final
// top-k (k <= 16).
#include "common.h"

#define QSA_TOPK_MAX 16
#define QSA_DOCS_PER_BLOCK 2048

__global__ void __launch_bounds__(256)
qsa_topk_stage1(const float* __restrict__ queries,  // [Q, D] L2-normalized
                const float* __restrict__ docs,     // [N, D] L2-normalized
                float* __restrict__ cand_scores,    // [Q, nblk, k]
                int* __restrict__ cand_ids,         // [Q, nblk, k]
                int Qn, int N, int D, int k) {
  const int qi = blockIdx.x;
  const int blk = blockIdx.y;
  const int nblk = gridDim.y;
  const int wave = threadIdx.x / QSA_WAVE;
  const int lane = threadIdx.x % QSA_WAVE;

  extern __shared__ __attribute__((aligned(16))) char smem_raw[];
  float* qsm = reinterpret_cast<float*>(smem_raw);          // [D]
  float* wsc = qsm + D;                                     // [4][k] scores
  int* wid = reinterpret_cast<int*>(wsc + 4 * QSA_TOPK_MAX);  // [4][k]

  for (int i = threadIdx.x; i < D; i += blockDim.x)
    qsm[i] = queries[(long long)qi * D + i];
  __syncthreads();

  // per-wave running top-k kept in lane-0 registers via ballot-free scheme:
  // each wave keeps its top-k in LDS (k small; one update per doc).
  if (lane < k) {
    wsc[wave * QSA_TOPK_MAX + lane] = -2.0f;
    wid[wave * QSA_TOPK_MAX + lane] = -1;
  }

  const int doc0 = blk * QSA_DOCS_PER_BLOCK;
  const int doc_end = min(N, doc0 + QSA_DOCS_PER_BLOCK);
  for (int doc = doc0 + wave; doc < doc_end; doc += 4) {
    const float* drow = docs + (long long)doc * D;
    float acc = 0.f;
    for (int i = lane; i < D; i += QSA_WAVE) acc = fmaf(drow[i], qsm[i], acc);
    const float score = wave_reduce_sum(acc);
    // lane 0 updates the wave's top-k (insertion into sorted-descending list)
    if (lane == 0) {
      float* ts = wsc + wave * QSA_TOPK_MAX;
      int* ti = wid + wave * QSA_TOPK_MAX;
      if (score > ts[k - 1]) {
        int j = k - 1;
        while (j > 0 && ts[j - 1] < score) {
          ts[j] = ts[j - 1];
          ti[j] = ti[j - 1];
          --j;
        }
        ts[j] = score;
        ti[j] = doc;
      }
    }
  }
  __syncthreads();
  // merge the 4 wave lists -> block top-k (thread 0; 4k items, k <= 16)
  if (threadIdx.x == 0) {
    float out_s[QSA_TOPK_MAX];
    int out_i[QSA_TOPK_MAX];
    for (int j = 0; j < k; ++j) { out_s[j] = -2.0f; out_i[j] = -1; }
    for (int w = 0; w < 4; ++w) {
      for (int j = 0; j < k; ++j) {
        const float sc = wsc[w * QSA_TOPK_MAX + j];
        const int id = wid[w * QSA_TOPK_MAX + j];
        if (sc > out_s[k - 1]) {
          int t = k - 1;
          while (t > 0 && out_s[t - 1] < sc) {
            out_s[t] = out_s[t - 1];
            out_i[t] = out_i[t - 1];
            --t;
          }
          out_s[t] = sc;
          out_i[t] = id;
        }
      }
    }
    float* cs = cand_scores + ((long long)qi * nblk + blk) * k;
    int* ci = cand_ids + ((long long)qi * nblk + blk) * k;
    for (int j = 0; j < k; ++j) { cs[j] = out_s[j]; ci[j] = out_i[j]; }
  }
}

__global__ void __launch_bounds__(64)
qsa_topk_stage2(const float* __restrict__ cand_scores,  // [Q, nblk, k]
                const int* __restrict__ cand_ids,
                float* __restrict__ out_scores,  // [Q, k]
                int* __restrict__ out_ids, int nblk, int k) {
  const int qi = blockIdx.x;
  if (threadIdx.x != 0) return;  // serial select over nblk*k (small)
  float out_s[QSA_TOPK_MAX];
  int out_i[QSA_TOPK_MAX];
  for (int j = 0; j < k; ++j) { out_s[j] = -2.0f; out_i[j] = -1; }
  const float* cs = cand_scores + (long long)qi * nblk * k;
  const int* ci = cand_ids + (long long)qi * nblk * k;
  for (int t = 0; t < nblk * k; ++t) {
    const float sc = cs[t];
    if (sc > out_s[k - 1]) {
      int j = k - 1;
      while (j > 0 && out_s[j - 1] < sc) {
        out_s[j] = out_s[j - 1];
        out_i[j] = out_i[j - 1];
        --j;
      }
      out_s[j] = sc;
      out_i[j] = ci[t];
    }
  }
  for (int j = 0; j < k; ++j) {
    out_scores[(long long)qi * k + j] = out_s[j];
    out_ids[(long long)qi * k + j] = out_i[j];
  }
}

extern "C" void qsa_topk_launch(const float* queries, const float* docs,
                                float* cand_scores, int* cand_ids,
                                float* out_scores, int* out_ids, int Q, int N,
                                int D, int k, int nblk, hipStream_t stream) {
  const size_t smem =
      (size_t)(D + 4 * QSA_TOPK_MAX) * sizeof(float) + 4 * QSA_TOPK_MAX * sizeof(int);
  hipLaunchKernelGGL(qsa_topk_stage1, dim3(Q, nblk), dim3(256), smem, stream,
                     queries, docs, cand_scores, cand_ids, Q, N, D, k);
  hipLaunchKernelGGL(qsa_topk_stage2, dim3(Q), dim3(64), 0, stream,
                     cand_scores, cand_ids, out_scores, out_ids, nblk, k);
}
