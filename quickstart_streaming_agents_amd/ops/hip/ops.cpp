// Torch bindings for the qsa MI355X (gfx950) kernels.  HIP-native: no CUDA
// naming, no compatibility shims — this extension only builds for ROCm.
// Operator inventory these kernels implement: SURVEY.md 2.4 K1-K10
// (embedding, cosine top-k, anomaly scoring, paged-attention LLM decode,
// window aggregation, Avro wire codec).
#include <hip/hip_runtime.h>
#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

#define CHK(x) TORCH_CHECK(x, #x)
#define CHK_DEV(t) TORCH_CHECK((t).is_cuda(), #t " must be on the GPU")
#define CHK_CONT(t) TORCH_CHECK((t).is_contiguous(), #t " must be contiguous")
#define CHK_BF16(t) TORCH_CHECK((t).scalar_type() == at::kBFloat16, #t " must be bf16")
#define CHK_F32(t) TORCH_CHECK((t).scalar_type() == at::kFloat, #t " must be f32")
#define CHK_I32(t) TORCH_CHECK((t).scalar_type() == at::kInt, #t " must be i32")

static inline const unsigned short* u16(const torch::Tensor& t) {
  return reinterpret_cast<const unsigned short*>(t.data_ptr());
}
static inline unsigned short* u16m(torch::Tensor& t) {
  return reinterpret_cast<unsigned short*>(t.data_ptr());
}
static inline hipStream_t cur_stream() {
  return at::hip::getCurrentHIPStream().stream();
}

// ---- launcher decls (defined in the .hip files) ---------------------------
extern "C" void qsa_rmsnorm_launch(const unsigned short*, const unsigned short*,
                                   unsigned short*, unsigned short*, long long,
                                   int, float, hipStream_t);
extern "C" void qsa_swiglu_launch(const unsigned short*, const unsigned short*,
                                  unsigned short*, long long, long long,
                                  long long, int, hipStream_t);
extern "C" void qsa_rope_launch(unsigned short*, unsigned short*, const float*,
                                const float*, const int*, int, int, int, int,
                                long long, long long, hipStream_t);
extern "C" void qsa_softmax_rows_launch(float*, int, int, int, int, int,
                                        const int*, hipStream_t);
extern "C" void qsa_softmax_rows_bf16_launch(unsigned short*, long long, int,
                                             float, const int*, hipStream_t);
extern "C" void qsa_paged_attn_mfma_launch(
    const unsigned short*, const unsigned short*, const unsigned short*,
    const int*, const int*, float*, float*, unsigned short*, float, int, int,
    int, int, int, long long, int, hipStream_t);
extern "C" void qsa_kv_append_launch(const unsigned short*,
                                     const unsigned short*, unsigned short*,
                                     unsigned short*, const int*, const int*,
                                     int, int, int, int, long long,
                                     hipStream_t);
extern "C" void qsa_paged_attn_prefill_launch(
    const unsigned short*, const unsigned short*, const unsigned short*,
    const int*, const int*, const int*, const int*, const int*, const int*,
    unsigned short*, float, int, int, int, int, int, long long, int,
    hipStream_t);
extern "C" void qsa_rope_kv_append_launch(unsigned short*,
                                          const unsigned short*,
                                          const unsigned short*,
                                          unsigned short*, unsigned short*,
                                          const float*, const float*,
                                          const int*, const int*, int, int,
                                          int, int, int, long long, long long,
                                          hipStream_t);
extern "C" void qsa_kv_scatter_launch(const unsigned short*,
                                      const unsigned short*, unsigned short*,
                                      unsigned short*, const int*, int, int,
                                      int, long long, hipStream_t);
extern "C" void qsa_skinny_gemm_launch(const unsigned short*,
                                       const unsigned short*, unsigned short*,
                                       int, int, long long, long long,
                                       hipStream_t);
extern "C" void qsa_skinny_gemm_probe_launch(const unsigned short*,
                                             const unsigned short*,
                                             unsigned short*, int, int,
                                             long long, long long, int, int,
                                             int, hipStream_t);
extern "C" void qsa_skinny_gemm_fp8_launch(const unsigned short*,
                                           const unsigned char*,
                                           const float*, unsigned short*,
                                           int, int, long long, long long,
                                           hipStream_t);
extern "C" void qsa_skinny_gemm_fp8_probe_launch(
    const unsigned short*, const unsigned char*, const float*,
    unsigned short*, int, int, long long, long long, int, int, int,
    hipStream_t);
extern "C" void qsa_gemm_fp8_batch_launch(const unsigned short*,
                                          const unsigned char*,
                                          const float*, unsigned short*,
                                          float*, int, int, long long,
                                          long long, int, hipStream_t);
extern "C" void qsa_hash_build_launch(const long long*, const long long*,
                                      unsigned long long*,
                                      unsigned long long*, int, unsigned int,
                                      hipStream_t);
extern "C" void qsa_hash_probe_launch(const long long*,
                                      const unsigned long long*,
                                      const unsigned long long*, int*, int,
                                      unsigned int, long long, hipStream_t);
extern "C" void qsa_topk_launch(const float*, const float*, float*, int*,
                                float*, int*, int, int, int, int, int,
                                hipStream_t);
extern "C" void qsa_window_agg_launch(const long long*, const int*,
                                      const float*, int*, float*, long long,
                                      long long, int, long long, int,
                                      hipStream_t);
void register_avro(pybind11::module_&);  // avro_codec.cpp
extern "C" void qsa_anomaly_batch_launch(const float*, const int*, float*,
                                         float*, int*, int, int, int,
                                         hipStream_t);

// ---------------------------------------------------------------------------

torch::Tensor rmsnorm(torch::Tensor x, torch::Tensor w, double eps) {
  CHK_DEV(x); CHK_CONT(x); CHK_BF16(x); CHK_BF16(w); CHK_CONT(w);
  const int H = x.size(-1);
  TORCH_CHECK(H % 8 == 0, "H % 8 == 0");
  const long long rows = x.numel() / H;
  auto y = torch::empty_like(x);
  qsa_rmsnorm_launch(u16(x), u16(w), u16m(y), nullptr, rows, H, (float)eps,
                     cur_stream());
  return y;
}

torch::Tensor rmsnorm_residual(torch::Tensor x, torch::Tensor res,
                               torch::Tensor w, double eps) {
  CHK_DEV(x); CHK_CONT(x); CHK_BF16(x); CHK_BF16(res); CHK_CONT(res);
  CHK_BF16(w); CHK_CONT(w);
  const int H = x.size(-1);
  TORCH_CHECK(H % 8 == 0, "H % 8 == 0");
  TORCH_CHECK(res.sizes() == x.sizes(), "residual shape mismatch");
  const long long rows = x.numel() / H;
  auto y = torch::empty_like(x);
  qsa_rmsnorm_launch(u16(x), u16(w), u16m(y), u16m(res), rows, H, (float)eps,
                     cur_stream());
  return y;
}

torch::Tensor swiglu(torch::Tensor gate, torch::Tensor up) {
  // gate/up: [rows, cols] views with equal row stride (e.g. halves of the
  // fused gate_up GEMM output); cols % 8 == 0, element-contiguous rows.
  CHK_DEV(gate); CHK_BF16(gate); CHK_BF16(up);
  TORCH_CHECK(gate.dim() == 2 && up.dim() == 2, "2-D");
  TORCH_CHECK(gate.sizes() == up.sizes(), "shape mismatch");
  TORCH_CHECK(gate.stride(1) == 1 && up.stride(1) == 1, "rows contiguous");
  TORCH_CHECK(gate.stride(0) == up.stride(0), "row strides must match");
  const long long rows = gate.size(0), cols = gate.size(1);
  TORCH_CHECK(cols % 8 == 0, "cols % 8 == 0");
  auto y = torch::empty({rows, cols}, gate.options());
  const long long n = rows * cols;
  const int blocks = (int)std::min<long long>((n / 8 + 255) / 256, 8192);
  qsa_swiglu_launch(u16(gate), u16(up), u16m(y), rows, cols, gate.stride(0),
                    blocks, cur_stream());
  return y;
}

static inline void chk_hd_strided(const torch::Tensor& t, int D,
                                  const char* name) {
  TORCH_CHECK(t.dim() == 3 && t.stride(2) == 1 && t.stride(1) == D,
              name, " must be a [B, H, D] view with contiguous heads");
}

void rope_inplace(torch::Tensor q, torch::Tensor k, torch::Tensor cos_t,
                  torch::Tensor sin_t, torch::Tensor pos) {
  CHK_DEV(q); CHK_BF16(q); CHK_BF16(k);
  CHK_F32(cos_t); CHK_F32(sin_t); CHK_I32(pos);
  const int B = q.size(0), QH = q.size(1), D = q.size(2);
  const int KVH = k.size(1);
  TORCH_CHECK(k.size(0) == B && k.size(2) == D, "k shape");
  chk_hd_strided(q, D, "q"); chk_hd_strided(k, D, "k");
  qsa_rope_launch(u16m(q), u16m(k), cos_t.data_ptr<float>(),
                  sin_t.data_ptr<float>(), pos.data_ptr<int>(), B, QH, KVH, D,
                  q.stride(0), k.stride(0), cur_stream());
}

void softmax_rows_(torch::Tensor scores, long col_offset, bool causal,
                   long row_mod, c10::optional<torch::Tensor> row_limits) {
  CHK_DEV(scores); CHK_CONT(scores); CHK_F32(scores);
  const int rows = scores.size(0), cols = scores.size(1);
  const int* rl = nullptr;
  if (row_limits.has_value()) {
    CHK_I32(row_limits.value());
    rl = row_limits.value().data_ptr<int>();
  }
  qsa_softmax_rows_launch(scores.data_ptr<float>(), rows, cols,
                          (int)col_offset, causal ? 1 : 0, (int)row_mod, rl,
                          cur_stream());
}

void softmax_rows_bf16_(torch::Tensor scores, double scale,
                        torch::Tensor row_limits) {
  CHK_DEV(scores); CHK_CONT(scores); CHK_BF16(scores); CHK_I32(row_limits);
  const long long rows = scores.size(0);
  const int cols = scores.size(1);
  TORCH_CHECK(cols % 8 == 0, "cols % 8 == 0");
  TORCH_CHECK(row_limits.numel() == rows, "one limit per row");
  qsa_softmax_rows_bf16_launch(u16m(scores), rows, cols, (float)scale,
                               row_limits.data_ptr<int>(), cur_stream());
}

torch::Tensor paged_attn_decode(torch::Tensor q, torch::Tensor kc,
                                torch::Tensor vc, torch::Tensor block_table,
                                torch::Tensor seq_lens, double scale) {
  CHK_DEV(q); CHK_BF16(q); CHK_BF16(kc); CHK_BF16(vc);
  CHK_CONT(kc); CHK_CONT(vc); CHK_I32(block_table); CHK_I32(seq_lens);
  CHK_CONT(block_table);
  const int B = q.size(0), QH = q.size(1), D = q.size(2);
  chk_hd_strided(q, D, "q");
  const int KVH = kc.size(1);
  TORCH_CHECK(D == 128 || D == 64, "D must be 64/128");
  TORCH_CHECK(QH % KVH == 0, "GQA requires QH % KVH == 0");
  TORCH_CHECK(QH / KVH <= 16, "GQA ratio <= 16");
  TORCH_CHECK(kc.size(2) == D / 8 && kc.size(3) == 64 && kc.size(4) == 8,
              "K cache layout [P, KVH, D/8, 64, 8]");
  TORCH_CHECK(vc.size(2) == D && vc.size(3) == 64,
              "V cache layout [P, KVH, D, 64] (transposed)");
  const int max_pages = block_table.size(1);
  auto out = torch::empty({B, QH, D}, q.options());
  // flash-decoding split count: fill the 256-CU chip (one WAVE per
  // (b, kvh, split); 4 splits share a workgroup)
  int ns = (int)std::min<long long>(32, std::max<long long>(
      1, (2048 + (long long)B * KVH - 1) / ((long long)B * KVH)));
  if (ns > 1) ns = std::max(ns, 4);   // fill all 4 waves per workgroup
  ns = std::min(ns, std::max(1, max_pages));
  auto opts_f = q.options().dtype(at::kFloat);
  auto part_o = torch::empty({B, QH, ns, D}, opts_f);
  auto part_ml = torch::empty({B, QH, ns, 2}, opts_f);
  qsa_paged_attn_mfma_launch(
      u16(q), u16(kc), u16(vc), block_table.data_ptr<int>(),
      seq_lens.data_ptr<int>(), part_o.data_ptr<float>(),
      part_ml.data_ptr<float>(), u16m(out), (float)scale, B, QH, KVH,
      max_pages, D, q.stride(0), ns, cur_stream());
  return out;
}

void kv_append(torch::Tensor knew, torch::Tensor vnew, torch::Tensor kc,
               torch::Tensor vc, torch::Tensor block_table,
               torch::Tensor seq_lens) {
  CHK_DEV(knew); CHK_BF16(knew); CHK_BF16(vnew);
  CHK_I32(block_table); CHK_I32(seq_lens);
  const int B = knew.size(0), KVH = knew.size(1), D = knew.size(2);
  chk_hd_strided(knew, D, "knew"); chk_hd_strided(vnew, D, "vnew");
  TORCH_CHECK(knew.stride(0) == vnew.stride(0), "k/v row strides must match");
  qsa_kv_append_launch(u16(knew), u16(vnew), u16m(kc), u16m(vc),
                       block_table.data_ptr<int>(), seq_lens.data_ptr<int>(),
                       B, KVH, D, (int)block_table.size(1), knew.stride(0),
                       cur_stream());
}

torch::Tensor paged_attn_prefill(torch::Tensor q, torch::Tensor kc,
                                 torch::Tensor vc, torch::Tensor block_table,
                                 torch::Tensor qb_item, torch::Tensor qb_pos0,
                                 torch::Tensor item_off,
                                 torch::Tensor item_start,
                                 torch::Tensor item_len, double scale,
                                 bool causal) {
  CHK_DEV(q); CHK_BF16(q); CHK_BF16(kc); CHK_BF16(vc);
  CHK_CONT(kc); CHK_CONT(vc); CHK_I32(block_table); CHK_CONT(block_table);
  CHK_I32(qb_item); CHK_I32(qb_pos0); CHK_I32(item_off);
  CHK_I32(item_start); CHK_I32(item_len);
  const int T = q.size(0), QH = q.size(1), D = q.size(2);
  chk_hd_strided(q, D, "q");
  const int KVH = kc.size(1);
  TORCH_CHECK(D == 128 || D == 64, "D must be 64/128");
  TORCH_CHECK(QH % KVH == 0 && QH / KVH <= 16, "GQA ratio <= 16");
  TORCH_CHECK(QH % 4 == 0, "QH % 4 == 0 (4 head-waves per workgroup "
              "lockstep on a shared page stream)");
  TORCH_CHECK(vc.size(2) == D && vc.size(3) == 64, "V layout [P,KVH,D,64]");
  const int QB = qb_item.numel();
  const int npmax = block_table.size(1);
  auto out = torch::empty({(long long)T, (long long)QH * D}, q.options());
  qsa_paged_attn_prefill_launch(
      u16(q), u16(kc), u16(vc), block_table.data_ptr<int>(),
      qb_item.data_ptr<int>(), qb_pos0.data_ptr<int>(),
      item_off.data_ptr<int>(), item_start.data_ptr<int>(),
      item_len.data_ptr<int>(), u16m(out), (float)scale, QB, QH, KVH, npmax,
      D, q.stride(0), causal ? 1 : 0, cur_stream());
  return out;
}

void rope_kv_append(torch::Tensor q, torch::Tensor k, torch::Tensor v,
                    torch::Tensor kc, torch::Tensor vc, torch::Tensor cos_t,
                    torch::Tensor sin_t, torch::Tensor block_table,
                    torch::Tensor seq_lens) {
  CHK_DEV(q); CHK_BF16(q); CHK_BF16(k); CHK_BF16(v);
  CHK_F32(cos_t); CHK_F32(sin_t); CHK_I32(block_table); CHK_I32(seq_lens);
  CHK_CONT(block_table);
  const int B = q.size(0), QH = q.size(1), D = q.size(2);
  const int KVH = k.size(1);
  chk_hd_strided(q, D, "q"); chk_hd_strided(k, D, "k");
  chk_hd_strided(v, D, "v");
  TORCH_CHECK(k.stride(0) == v.stride(0), "k/v strides must match");
  qsa_rope_kv_append_launch(u16m(q), u16(k), u16(v), u16m(kc), u16m(vc),
                            cos_t.data_ptr<float>(), sin_t.data_ptr<float>(),
                            block_table.data_ptr<int>(),
                            seq_lens.data_ptr<int>(), B, QH, KVH, D,
                            (int)block_table.size(1), q.stride(0),
                            k.stride(0), cur_stream());
}

void kv_scatter(torch::Tensor knew, torch::Tensor vnew, torch::Tensor kc,
                torch::Tensor vc, torch::Tensor slots) {
  CHK_DEV(knew); CHK_BF16(knew); CHK_BF16(vnew); CHK_I32(slots);
  const int T = knew.size(0), KVH = knew.size(1), D = knew.size(2);
  chk_hd_strided(knew, D, "knew"); chk_hd_strided(vnew, D, "vnew");
  TORCH_CHECK(knew.stride(0) == vnew.stride(0), "k/v strides must match");
  if (T == 0) return;
  qsa_kv_scatter_launch(u16(knew), u16(vnew), u16m(kc), u16m(vc),
                        slots.data_ptr<int>(), T, KVH, D, knew.stride(0),
                        cur_stream());
}

torch::Tensor pack_weight_frag(torch::Tensor w) {
  // [N, K] bf16 -> MFMA-fragment-major [N/16, K/32, 16, 32] for the
  // skinny-GEMM weight stream (one wave reads one contiguous 1 KiB block).
  CHK_BF16(w); CHK_CONT(w);
  const long long N = w.size(0), K = w.size(1);
  TORCH_CHECK(N % 16 == 0 && K % 32 == 0, "pack needs N%16==0, K%32==0");
  return w.reshape({N / 16, 16, K / 32, 32})
      .permute({0, 2, 1, 3}).contiguous();
}

torch::Tensor skinny_gemm(torch::Tensor a, torch::Tensor wf, long N, long K) {
  // C[M,N] = a[M,K] @ W^T with W pre-packed by pack_weight_frag.
  CHK_DEV(a); CHK_BF16(a); CHK_BF16(wf); CHK_CONT(wf);
  TORCH_CHECK(a.dim() == 2 && a.stride(1) == 1, "a rows must be contiguous");
  const int M = a.size(0);
  TORCH_CHECK(M >= 1 && M <= 32, "skinny_gemm: M in [1,32]");
  TORCH_CHECK(a.size(1) == K, "K mismatch");
  TORCH_CHECK(K % 256 == 0 && N % 16 == 0, "K%256==0, N%16==0");
  TORCH_CHECK(wf.numel() == (long long)N * K, "wf size");
  auto out = torch::empty({M, (long long)N}, a.options());
  qsa_skinny_gemm_launch(u16(a), u16(wf), u16m(out), M, (int)N, K,
                         a.stride(0), cur_stream());
  return out;
}

torch::Tensor skinny_gemm_probe(torch::Tensor a, torch::Tensor wf, long N,
                                long K, long waves, long nt, long variant) {
  CHK_DEV(a); CHK_BF16(a); CHK_BF16(wf); CHK_CONT(wf);
  const int M = a.size(0);
  auto out = torch::empty({M, (long long)N}, a.options());
  qsa_skinny_gemm_probe_launch(u16(a), u16(wf), u16m(out), M, (int)N, K,
                               a.stride(0), (int)waves, (int)nt,
                               (int)variant, cur_stream());
  return out;
}

torch::Tensor gemm_fp8_batch(torch::Tensor a, torch::Tensor qf,
                             torch::Tensor scale, long N, long K,
                             long splitk) {
  // C[M,N] = a @ (scale * dequant(Q))^T for decode batches 32 < M <= 256
  CHK_DEV(a); CHK_BF16(a); CHK_CONT(qf);
  TORCH_CHECK(qf.scalar_type() == torch::kUInt8, "qf must be uint8 (fp8)");
  TORCH_CHECK(scale.scalar_type() == torch::kFloat32 && scale.is_cuda() &&
                  scale.is_contiguous() && scale.numel() == N, "scale");
  TORCH_CHECK(a.dim() == 2 && a.stride(1) == 1, "a rows must be contiguous");
  const int M = a.size(0);
  TORCH_CHECK(M >= 1 && M <= 256, "gemm_fp8_batch: M in [1,256]");
  TORCH_CHECK(a.size(1) == K, "K mismatch");
  TORCH_CHECK(N % 64 == 0, "N % 64 == 0");
  TORCH_CHECK(splitk == 1 || splitk == 2 || splitk == 4, "splitk 1/2/4");
  TORCH_CHECK(K % (64 * splitk) == 0, "K % (64*splitk) == 0");
  TORCH_CHECK(qf.numel() == (long long)N * K, "qf size");
  auto out = torch::empty({(long long)M, (long long)N}, a.options());
  torch::Tensor ws;
  float* wsp = nullptr;
  if (splitk > 1) {
    ws = torch::empty({splitk, (long long)M, (long long)N},
                      a.options().dtype(torch::kFloat32));
    wsp = ws.data_ptr<float>();
  }
  qsa_gemm_fp8_batch_launch(u16(a), qf.data_ptr<unsigned char>(),
                            scale.data_ptr<float>(), u16m(out), wsp, M,
                            (int)N, K, a.stride(0), (int)splitk,
                            cur_stream());
  return out;
}

std::vector<torch::Tensor> hash_build(torch::Tensor keys,
                                      torch::Tensor ts) {
  // latest-event-time-per-key hash table in HBM (K8 streaming join state)
  CHK_DEV(keys);
  TORCH_CHECK(keys.scalar_type() == torch::kInt64 && keys.is_contiguous(),
              "keys must be contiguous i64");
  TORCH_CHECK(ts.scalar_type() == torch::kInt64 && ts.is_contiguous() &&
                  ts.numel() == keys.numel(), "ts must be i64 like keys");
  const long long n = keys.numel();
  TORCH_CHECK(n < (1 << 24), "hash_build: <= 2^24 rows per batch");
  long long cap = 64;
  while (cap < 2 * n) cap <<= 1;
  auto opts = keys.options();
  auto tkeys = torch::full({cap}, -1, opts);   // all-ones bits == EMPTY
  auto tpay = torch::zeros({cap}, opts);
  qsa_hash_build_launch((const long long*)keys.data_ptr<int64_t>(),
                        (const long long*)ts.data_ptr<int64_t>(),
                        (unsigned long long*)tkeys.data_ptr<int64_t>(),
                        (unsigned long long*)tpay.data_ptr<int64_t>(),
                        (int)n, (unsigned int)(cap - 1), cur_stream());
  return {tkeys, tpay};
}

torch::Tensor hash_probe(torch::Tensor tkeys, torch::Tensor tpay,
                         torch::Tensor keys, long long min_ts) {
  CHK_DEV(keys);
  TORCH_CHECK(keys.scalar_type() == torch::kInt64 && keys.is_contiguous(),
              "keys must be contiguous i64");
  const long long m = keys.numel();
  const long long cap = tkeys.numel();
  TORCH_CHECK((cap & (cap - 1)) == 0, "table capacity must be pow2");
  auto out = torch::empty({m}, keys.options().dtype(torch::kInt32));
  qsa_hash_probe_launch((const long long*)keys.data_ptr<int64_t>(),
                        (const unsigned long long*)tkeys.data_ptr<int64_t>(),
                        (const unsigned long long*)tpay.data_ptr<int64_t>(),
                        out.data_ptr<int>(), (int)m,
                        (unsigned int)(cap - 1), min_ts, cur_stream());
  return out;
}

torch::Tensor skinny_gemm_fp8(torch::Tensor a, torch::Tensor qf,
                              torch::Tensor scale, long N, long K) {
  // C[M,N] = a @ (scale[N] * dequant(Q))^T; Q pre-packed fp8 e4m3 in the
  // fragment-pair-major stream layout (skinny_gemm_fp8.hip header).
  CHK_DEV(a); CHK_BF16(a); CHK_CONT(qf);
  TORCH_CHECK(qf.scalar_type() == torch::kUInt8, "qf must be uint8 (fp8)");
  TORCH_CHECK(scale.scalar_type() == torch::kFloat32 && scale.is_cuda() &&
                  scale.is_contiguous() && scale.numel() == N,
              "scale must be f32 [N] on device");
  TORCH_CHECK(a.dim() == 2 && a.stride(1) == 1, "a rows must be contiguous");
  const int M = a.size(0);
  TORCH_CHECK(M >= 1 && M <= 32, "skinny_gemm_fp8: M in [1,32]");
  TORCH_CHECK(a.size(1) == K, "K mismatch");
  TORCH_CHECK(K % 256 == 0 && N % 16 == 0, "K%256==0, N%16==0");
  TORCH_CHECK(qf.numel() == (long long)N * K, "qf size");
  auto out = torch::empty({M, (long long)N}, a.options());
  qsa_skinny_gemm_fp8_launch(u16(a), qf.data_ptr<unsigned char>(),
                             scale.data_ptr<float>(), u16m(out), M, (int)N,
                             K, a.stride(0), cur_stream());
  return out;
}

torch::Tensor skinny_gemm_fp8_probe(torch::Tensor a, torch::Tensor qf,
                                    torch::Tensor scale, long N, long K,
                                    long waves, long tiles, long nt) {
  CHK_DEV(a); CHK_BF16(a); CHK_CONT(qf);
  const int M = a.size(0);
  auto out = torch::empty({M, (long long)N}, a.options());
  qsa_skinny_gemm_fp8_probe_launch(u16(a), qf.data_ptr<unsigned char>(),
                                   scale.data_ptr<float>(), u16m(out), M,
                                   (int)N, K, a.stride(0), (int)waves,
                                   (int)tiles, (int)nt, cur_stream());
  return out;
}

std::vector<torch::Tensor> topk_cosine(torch::Tensor queries,
                                       torch::Tensor docs, long k) {
  CHK_DEV(queries); CHK_CONT(queries); CHK_F32(queries); CHK_F32(docs);
  CHK_CONT(docs);
  const int Q = queries.size(0), D = queries.size(1), N = docs.size(0);
  TORCH_CHECK(docs.size(1) == D, "dim mismatch");
  TORCH_CHECK(k >= 1 && k <= 16, "k in [1,16]");
  const int nblk = std::max(1, (N + 2047) / 2048);
  auto opts_f = queries.options();
  auto opts_i = queries.options().dtype(at::kInt);
  auto cand_s = torch::empty({Q, nblk, k}, opts_f);
  auto cand_i = torch::empty({Q, nblk, k}, opts_i);
  auto out_s = torch::empty({Q, k}, opts_f);
  auto out_i = torch::empty({Q, k}, opts_i);
  qsa_topk_launch(queries.data_ptr<float>(), docs.data_ptr<float>(),
                  cand_s.data_ptr<float>(), cand_i.data_ptr<int>(),
                  out_s.data_ptr<float>(), out_i.data_ptr<int>(), Q, N, D,
                  (int)k, nblk, cur_stream());
  return {out_s, out_i};
}

std::vector<torch::Tensor> window_agg(torch::Tensor ts, torch::Tensor key,
                                      c10::optional<torch::Tensor> value,
                                      long t0, long win_ms, long nwin,
                                      long nkeys) {
  CHK_DEV(ts); CHK_CONT(ts); CHK_I32(key);
  TORCH_CHECK(ts.scalar_type() == at::kLong, "ts must be i64");
  const long long n = ts.numel();
  auto counts = torch::zeros({nkeys, nwin}, key.options());
  auto sums = torch::zeros({nkeys, nwin}, ts.options().dtype(at::kFloat));
  const float* vptr = nullptr;
  if (value.has_value()) {
    CHK_F32(value.value());
    vptr = value.value().data_ptr<float>();
  }
  const int blocks = (int)std::min<long long>((n + 255) / 256, 4096);
  qsa_window_agg_launch(reinterpret_cast<const long long*>(ts.data_ptr<int64_t>()),
                        key.data_ptr<int>(), vptr,
                        counts.data_ptr<int>(), sums.data_ptr<float>(), t0,
                        win_ms, (int)nwin, n, std::max(blocks, 1),
                        cur_stream());
  return {counts, sums};
}

std::vector<torch::Tensor> anomaly_batch(torch::Tensor series,
                                         torch::Tensor lengths, long order) {
  CHK_DEV(series); CHK_CONT(series); CHK_F32(series); CHK_I32(lengths);
  const int K = series.size(0), Tmax = series.size(1);
  auto fc = torch::empty({K}, series.options());
  auto se = torch::empty({K}, series.options());
  auto dof = torch::empty({K}, series.options().dtype(at::kInt));
  qsa_anomaly_batch_launch(series.data_ptr<float>(), lengths.data_ptr<int>(),
                           fc.data_ptr<float>(), se.data_ptr<float>(),
                           dof.data_ptr<int>(), K, Tmax, (int)order,
                           cur_stream());
  return {fc, se, dof};
}

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
  m.def("rmsnorm", &rmsnorm, "fused RMSNorm (bf16)");
  m.def("rmsnorm_residual", &rmsnorm_residual,
        "fused residual-add + RMSNorm (bf16; residual updated in place)");
  m.def("swiglu", &swiglu, "fused silu(gate)*up (bf16)");
  m.def("rope_inplace", &rope_inplace, "rotary embedding in place (bf16)");
  m.def("softmax_rows_bf16_", &softmax_rows_bf16_,
        "bf16 masked row softmax with folded scale (in place)");
  m.def("softmax_rows_", &softmax_rows_, "row softmax in place (f32)",
        py::arg("scores"), py::arg("col_offset") = 0, py::arg("causal") = false,
        py::arg("row_mod") = 0, py::arg("row_limits") = py::none());
  m.def("paged_attn_prefill", &paged_attn_prefill,
        "varlen flash prefill attention over the paged cache");
  m.def("paged_attn_decode", &paged_attn_decode,
        "paged-attention decode (bf16, GQA, page=64)");
  m.def("kv_append", &kv_append, "append one step's k/v to the paged cache");
  m.def("rope_kv_append", &rope_kv_append,
        "fused decode rope + paged-cache append");
  m.def("kv_scatter", &kv_scatter, "scatter prefill k/v by slot ids");
  m.def("pack_weight_frag", &pack_weight_frag,
        "repack [N,K] bf16 into MFMA-fragment-major for skinny_gemm");
  m.def("skinny_gemm", &skinny_gemm,
        "decode-batch GEMM (M<=32) on the packed weight stream");
  m.def("skinny_gemm_probe", &skinny_gemm_probe,
        "ablation probe: waves/nt/variant sweep");
  m.def("skinny_gemm_fp8", &skinny_gemm_fp8,
        "fp8-weight decode GEMM (M<=32): half the weight stream");
  m.def("gemm_fp8_batch", &gemm_fp8_batch,
        "batched-M (<=256) fp8 weight-stream GEMM with optional split-K");
  m.def("skinny_gemm_fp8_probe", &skinny_gemm_fp8_probe,
        "fp8 ablation probe: waves/tiles/nt sweep");
  m.def("topk_cosine", &topk_cosine, "exact cosine top-k over the HBM index");
  m.def("hash_build", &hash_build,
        "build latest-per-key hash table from (keys, event ts)");
  m.def("hash_probe", &hash_probe,
        "probe the hash table with TTL cutoff -> row indices or -1");
  m.def("window_agg", &window_agg, "segmented (key, window) count/sum");
  m.def("anomaly_batch", &anomaly_batch, "batched AR+ridge anomaly scorer");
  register_avro(m);
}
