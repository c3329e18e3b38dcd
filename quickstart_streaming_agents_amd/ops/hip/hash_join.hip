// GPU hash join for streaming enrichment (K8):
// build a latest-row-per-key hash table from a dimension batch in HBM,
// probe it with a stream batch, TTL-filter at probe time.
//
// Semantics match runtime/joins.TTLTable + enrich_join (the reference's
// `orders JOIN customers JOIN products` under SET 'sql.state-ttl',
// LAB1-Walkthrough.md:119-131): keys are pre-hashed to i64 on the host
// (columnar batches from the Avro codec), the table keeps the row with
// the LATEST event time per key, and a probe returns that row's index or
// -1 if absent/expired.
//
// Table layout: open addressing, linear probing, power-of-two capacity.
//   keys[cap]  : u64, EMPTY = 0xFFFFFFFFFFFFFFFF (hashed keys never
//                collide with it: host masks hashes to 63 bits)
//   pay[cap]   : u64 packed (ts << 24) | row  — one atomicMax keeps the
//                latest-ts row per key (row < 2^24, ts < 2^40 ms ≈ 34 y)
// Insert: linear-probe atomicCAS on the key slot, then atomicMax on the
// payload.  No locks, no host round trips; eviction is at probe time
// (ts >= cutoff), full compaction = rebuild from the surviving rows.
#include "common.h"

#define QSA_HJ_EMPTY 0xFFFFFFFFFFFFFFFFull

__device__ __forceinline__ unsigned long long qsa_hj_mix(
    unsigned long long x) {
  // splitmix64 finalizer: spreads host hashes over the table
  x += 0x9E3779B97F4A7C15ull;
  x = (x ^ (x >> 30)) * 0xBF58476D1CE4E5B9ull;
  x = (x ^ (x >> 27)) * 0x94D049BB133111EBull;
  return x ^ (x >> 31);
}

__global__ void qsa_hash_build(const long long* __restrict__ keys,   // [N]
                               const long long* __restrict__ ts,     // [N]
                               unsigned long long* __restrict__ tkeys,
                               unsigned long long* __restrict__ tpay,
                               int n, unsigned int mask) {
  const int i = blockIdx.x * blockDim.x + threadIdx.x;
  if (i >= n) return;
  const unsigned long long key = (unsigned long long)keys[i];
  const unsigned long long pay =
      (((unsigned long long)ts[i]) << 24) | (unsigned int)(i & 0xFFFFFF);
  unsigned int slot = (unsigned int)qsa_hj_mix(key) & mask;
  for (unsigned int probe = 0; probe <= mask; ++probe) {
    unsigned long long prev =
        atomicCAS(&tkeys[slot], QSA_HJ_EMPTY, key);
    if (prev == QSA_HJ_EMPTY || prev == key) {
      atomicMax(&tpay[slot], pay);   // latest event time wins
      return;
    }
    slot = (slot + 1) & mask;
  }
  // table full: drop (host sizes cap >= 2*n, unreachable)
}

__global__ void qsa_hash_probe(const long long* __restrict__ keys,   // [M]
                               const unsigned long long* __restrict__ tkeys,
                               const unsigned long long* __restrict__ tpay,
                               int* __restrict__ out_row,            // [M]
                               int m, unsigned int mask,
                               long long min_ts) {
  const int i = blockIdx.x * blockDim.x + threadIdx.x;
  if (i >= m) return;
  const unsigned long long key = (unsigned long long)keys[i];
  unsigned int slot = (unsigned int)qsa_hj_mix(key) & mask;
  int row = -1;
  for (unsigned int probe = 0; probe <= mask; ++probe) {
    const unsigned long long k = tkeys[slot];
    if (k == QSA_HJ_EMPTY) break;
    if (k == key) {
      const unsigned long long pay = tpay[slot];
      const long long rts = (long long)(pay >> 24);
      if (rts >= min_ts) row = (int)(pay & 0xFFFFFF);
      break;
    }
    slot = (slot + 1) & mask;
  }
  out_row[i] = row;
}

extern "C" void qsa_hash_build_launch(const long long* keys,
                                      const long long* ts,
                                      unsigned long long* tkeys,
                                      unsigned long long* tpay, int n,
                                      unsigned int mask,
                                      hipStream_t stream) {
  if (n == 0) return;
  const int bs = 256;
  hipLaunchKernelGGL(qsa_hash_build, dim3((n + bs - 1) / bs), dim3(bs), 0,
                     stream, keys, ts, tkeys, tpay, n, mask);
}

extern "C" void qsa_hash_probe_launch(const long long* keys,
                                      const unsigned long long* tkeys,
                                      const unsigned long long* tpay,
                                      int* out_row, int m, unsigned int mask,
                                      long long min_ts, hipStream_t stream) {
  if (m == 0) return;
  const int bs = 256;
  hipLaunchKernelGGL(qsa_hash_probe, dim3((m + bs - 1) / bs), dim3(bs), 0,
                     stream, keys, tkeys, tpay, out_row, m, mask, min_ts);
}
