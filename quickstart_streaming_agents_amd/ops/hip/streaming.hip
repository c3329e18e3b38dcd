// Streaming-analytics kernels: batched windowed aggregation (K7) and the
// batched per-key AR anomaly scorer (K3).
#include "common.h"

// ---------------------------------------------------------------------------
// Windowed aggregation: segmented count/sum by (key, window) over an event
// batch.  window = (ts - t0) / win_ms; out slot = key * nwin + window.
// The reference shape: TUMBLE(...) GROUP BY key (LAB3:99-133, LAB4:127-141).
// ---------------------------------------------------------------------------
__global__ void __launch_bounds__(256)
qsa_window_agg(const long long* __restrict__ ts, const int* __restrict__ key,
               const float* __restrict__ value,  // nullptr -> count only
               int* __restrict__ counts, float* __restrict__ sums,
               long long t0, long long win_ms, int nwin, long long n) {
  long long i = (long long)blockIdx.x * blockDim.x + threadIdx.x;
  const long long stride = (long long)gridDim.x * blockDim.x;
  for (; i < n; i += stride) {
    const long long w = (ts[i] - t0) / win_ms;
    if (w < 0 || w >= nwin) continue;
    const long long slot = (long long)key[i] * nwin + w;
    atomicAdd(counts + slot, 1);
    if (value && sums) atomicAdd(sums + slot, value[i]);
  }
}

// ---------------------------------------------------------------------------
// Batched AR(p)+ridge anomaly scorer: one wave (64 lanes) per key series.
// Mirrors runtime/anomaly.py ar_forecast(): centered lag design, ridge
// lambda = 0.3 * tr(G)/p, prediction se with leverage.  p <= 4.
// series: [K, Tmax] f32 (ragged via lengths[]); outputs per key.
// ---------------------------------------------------------------------------
#define QSA_AR_MAXP 4

__global__ void __launch_bounds__(64)
qsa_anomaly_batch(const float* __restrict__ series, const int* __restrict__ lengths,
                  float* __restrict__ forecast, float* __restrict__ pred_se,
                  int* __restrict__ dof_out, int Tmax, int order) {
  const int kidx = blockIdx.x;
  const int lane = threadIdx.x;
  const float* h = series + (long long)kidx * Tmax;
  const int n = lengths[kidx];

  if (n < 6) {  // mean/std fallback (matches CPU path)
    float sum = 0.f, sq = 0.f;
    for (int i = lane; i < n; i += QSA_WAVE) {
      sum += h[i];
      sq = fmaf(h[i], h[i], sq);
    }
    sum = wave_reduce_sum(sum);
    sq = wave_reduce_sum(sq);
    if (lane == 0) {
      const float mean = n > 0 ? sum / n : 0.f;
      float sd = n > 1 ? sqrtf(fmaxf((sq - n * mean * mean) / (n - 1), 0.f))
                       : fabsf(mean) + 1.f;
      forecast[kidx] = mean;
      pred_se[kidx] = fmaxf(sd, 1e-9f);
      dof_out[kidx] = max(n - 1, 1);
    }
    return;
  }

  int p = min(order, max(1, (n - 4) / 4));
  p = min(p, QSA_AR_MAXP);
  const int m = n - p;

  // accumulate: colsum[j] (lag j+1), ysum, then centered cross products.
  float colsum[QSA_AR_MAXP + 1];  // [0] = y
#pragma unroll
  for (int j = 0; j <= QSA_AR_MAXP; ++j) colsum[j] = 0.f;
  for (int t = p + lane; t < n; t += QSA_WAVE) {
    colsum[0] += h[t];
#pragma unroll
    for (int j = 1; j <= QSA_AR_MAXP; ++j)
      if (j <= p) colsum[j] += h[t - j];
  }
#pragma unroll
  for (int j = 0; j <= QSA_AR_MAXP; ++j) colsum[j] = wave_reduce_sum(colsum[j]);
  const float ym = colsum[0] / m;
  float xm[QSA_AR_MAXP];
#pragma unroll
  for (int j = 0; j < QSA_AR_MAXP; ++j) xm[j] = (j < p) ? colsum[j + 1] / m : 0.f;

  // G = Xc'Xc (p x p), b = Xc'yc
  float G[QSA_AR_MAXP][QSA_AR_MAXP], bb[QSA_AR_MAXP];
#pragma unroll
  for (int a = 0; a < QSA_AR_MAXP; ++a) {
    bb[a] = 0.f;
#pragma unroll
    for (int c = 0; c < QSA_AR_MAXP; ++c) G[a][c] = 0.f;
  }
  for (int t = p + lane; t < n; t += QSA_WAVE) {
    float xc[QSA_AR_MAXP];
#pragma unroll
    for (int j = 0; j < QSA_AR_MAXP; ++j)
      xc[j] = (j < p) ? h[t - 1 - j] - xm[j] : 0.f;
    const float yc = h[t] - ym;
#pragma unroll
    for (int a = 0; a < QSA_AR_MAXP; ++a) {
      bb[a] = fmaf(xc[a], yc, bb[a]);
#pragma unroll
      for (int c = 0; c < QSA_AR_MAXP; ++c)
        G[a][c] = fmaf(xc[a], xc[c], G[a][c]);
    }
  }
#pragma unroll
  for (int a = 0; a < QSA_AR_MAXP; ++a) {
    bb[a] = wave_reduce_sum(bb[a]);
#pragma unroll
    for (int c = 0; c < QSA_AR_MAXP; ++c) G[a][c] = wave_reduce_sum(G[a][c]);
  }

  // lane 0: ridge solve (G + lam I) coef = b, Gauss-Jordan on p<=4;
  // also invert for the leverage term.
  if (lane == 0) {
    float tr = 0.f;
    for (int a = 0; a < p; ++a) tr += G[a][a];
    const float lam = 0.3f * (tr / p + 1e-12f);
    // augmented [A | I | b]
    float A[QSA_AR_MAXP][2 * QSA_AR_MAXP + 1];
    for (int a = 0; a < p; ++a) {
      for (int c = 0; c < p; ++c) A[a][c] = G[a][c] + (a == c ? lam : 0.f);
      for (int c = 0; c < p; ++c) A[a][p + c] = (a == c) ? 1.f : 0.f;
      A[a][2 * p] = bb[a];
    }
    for (int col = 0; col < p; ++col) {
      // partial pivot
      int piv = col;
      for (int rr2 = col + 1; rr2 < p; ++rr2)
        if (fabsf(A[rr2][col]) > fabsf(A[piv][col])) piv = rr2;
      if (piv != col)
        for (int c = 0; c <= 2 * p; ++c) {
          float tmp = A[col][c]; A[col][c] = A[piv][c]; A[piv][c] = tmp;
        }
      const float d = A[col][col];
      const float dinv = (fabsf(d) > 1e-30f) ? 1.f / d : 0.f;
      for (int c = 0; c <= 2 * p; ++c) A[col][c] *= dinv;
      for (int rr2 = 0; rr2 < p; ++rr2) {
        if (rr2 == col) continue;
        const float f = A[rr2][col];
        for (int c = 0; c <= 2 * p; ++c) A[rr2][c] -= f * A[col][c];
      }
    }
    float coef[QSA_AR_MAXP];
    for (int a = 0; a < p; ++a) coef[a] = A[a][2 * p];
    // residual variance: (yc'yc - 2 coef'b + coef'G coef) / dof
    // recompute yc'yc quickly (lane 0 serial over n-p; n <= 7000 ok)
    float yy = 0.f;
    for (int t = p; t < n; ++t) {
      const float yc = h[t] - ym;
      float pred = 0.f;
      for (int j = 0; j < p; ++j) pred = fmaf(coef[j], h[t - 1 - j] - xm[j], pred);
      const float r = yc - pred;
      yy = fmaf(r, r, yy);
    }
    const int dof = max(m - (p + 1), 1);
    const float resid_var = yy / dof;
    // x_next, leverage via inverse block
    float xn[QSA_AR_MAXP];
    for (int j = 0; j < p; ++j) xn[j] = h[n - 1 - j] - xm[j];
    float fc = ym;
    for (int j = 0; j < p; ++j) fc = fmaf(coef[j], xn[j], fc);
    float lever = 1.f / m;
    for (int a = 0; a < p; ++a) {
      float tmp = 0.f;
      for (int c = 0; c < p; ++c) tmp = fmaf(A[a][p + c], xn[c], tmp);
      lever = fmaf(xn[a], tmp, lever);
    }
    forecast[kidx] = fc;
    pred_se[kidx] =
        fmaxf(sqrtf(fmaxf(resid_var, 1e-18f) * (1.f + fmaxf(lever, 0.f))),
              1e-9f);
    dof_out[kidx] = dof;
  }
}

extern "C" void qsa_window_agg_launch(const long long* ts, const int* key,
                                      const float* value, int* counts,
                                      float* sums, long long t0,
                                      long long win_ms, int nwin, long long n,
                                      int blocks, hipStream_t stream) {
  hipLaunchKernelGGL(qsa_window_agg, dim3(blocks), dim3(256), 0, stream, ts,
                     key, value, counts, sums, t0, win_ms, nwin, n);
}

extern "C" void qsa_anomaly_batch_launch(const float* series,
                                         const int* lengths, float* forecast,
                                         float* pred_se, int* dof_out, int K,
                                         int Tmax, int order,
                                         hipStream_t stream) {
  hipLaunchKernelGGL(qsa_anomaly_batch, dim3(K), dim3(64), 0, stream, series,
                     lengths, forecast, pred_se, dof_out, Tmax, order);
}
