// Python bindings for the CDNA4 HIP kernel library (substratus_amd._C).
// Validation lives here; kernels are thin and trusting.
#include <torch/extension.h>
#include <ATen/cuda/CUDAContext.h>
#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>

typedef __hip_bfloat16 bf16_t;

// launchers (defined in the .hip translation units)
void launch_rms_norm(bf16_t*, const bf16_t*, const bf16_t*, long, int, float, hipStream_t);
void launch_rms_norm_residual(bf16_t*, bf16_t*, const bf16_t*, const bf16_t*, long, int, float, hipStream_t);
void launch_layer_norm(bf16_t*, const bf16_t*, const bf16_t*, const bf16_t*, long, int, float, hipStream_t);
void launch_silu_mul(bf16_t*, const bf16_t*, long, int, hipStream_t);
void launch_gelu(bf16_t*, const bf16_t*, long, hipStream_t);
void launch_rope_kv_append(bf16_t*, bf16_t*, const bf16_t*, const int*, void*, void*, const int*, int, int, int, int, int, float, bool, long, long, long, hipStream_t);
void launch_paged_attn_decode(bf16_t*, const bf16_t*, const void*, const void*, const int*, const int*, int, int, int, int, int, int, float, long, int, float*, float*, bool, hipStream_t);
int paged_attn_nsplit(int, int, int, int);
bool paged_attn_decode_supported(int, int);
void launch_attn_prefill(bf16_t*, const bf16_t*, const bf16_t*, const bf16_t*, const int*, int, int, int, int, int, float, hipStream_t, float*);
bool attn_prefill_supported(int);
void launch_attn_v5(bf16_t*, const bf16_t*, const bf16_t*, const bf16_t*, const int*, int, int, int, int, int, float, hipStream_t, float*);
bool attn_v5_supported(int);
void launch_attn_bwd_dq_v5(bf16_t*, const bf16_t*, const bf16_t*, const bf16_t*, const bf16_t*, const float*, const float*, const int*, int, int, int, int, int, float, hipStream_t);
void launch_attn_bwd_dkv_v5(float*, float*, const bf16_t*, const bf16_t*, const bf16_t*, const bf16_t*, const float*, const float*, const int*, int, int, int, int, int, float, hipStream_t);
void launch_attn_bwd_dkv_only(float*, float*, const bf16_t*, const bf16_t*, const bf16_t*, const bf16_t*, const float*, const float*, const int*, int, int, int, int, int, float, hipStream_t);
void launch_attn_bwd_delta(float*, const bf16_t*, const bf16_t*, long, int, hipStream_t);
void launch_attn_bwd(bf16_t*, float*, float*, const bf16_t*, const bf16_t*, const bf16_t*, const bf16_t*, const float*, const float*, const int*, int, int, int, int, int, float, hipStream_t);
bool attn_bwd_supported(int);
void launch_attn_prefill_probe(int, bf16_t*, const bf16_t*, const bf16_t*, const bf16_t*, const int*, int, int, int, int, float, hipStream_t);
void launch_mfma_probe(float*, const bf16_t*, const bf16_t*, hipStream_t);
void launch_mfma32_probe(float*, const bf16_t*, const bf16_t*, hipStream_t);
void launch_softmax_sample(long*, const bf16_t*, const float*, const int*, const float*, const float*, int, int, hipStream_t);
void launch_cross_entropy(float*, bf16_t*, const bf16_t*, const long*, long, int, long, float, hipStream_t);
void launch_adamw(bf16_t*, const bf16_t*, float*, float*, long, float, float, float, float, float, int, float, hipStream_t);
void launch_adamw_f32(float*, const float*, float*, float*, long, float, float, float, float, float, int, float, hipStream_t);
void launch_quant_gemv(bf16_t*, const bf16_t*, const void*, const unsigned char*, int, int, int, bool, hipStream_t);
void launch_dequant(bf16_t*, const void*, const unsigned char*, long, int, bool, hipStream_t);
void launch_quant_gemv_k(bf16_t*, const bf16_t*, const void*, const unsigned char*, int, int, int, bool, hipStream_t);
void launch_dequant_k(bf16_t*, const void*, const unsigned char*, int, int, bool, hipStream_t);
int decode_gemm_splitk(int, int);
bool decode_gemm_supported(int, int, int);
void launch_decode_gemm(bf16_t*, const bf16_t*, const bf16_t*, float*, int, int, int, int, hipStream_t);
int mfma_gemm_splitk(int, int, int);
bool mfma_gemm_supported(int, int, int);
void launch_mfma_gemm(bf16_t*, const bf16_t*, const bf16_t*, const bf16_t*, float*, int, int, int, int, hipStream_t);
int wstream_splitk(int, int);
bool wstream_supported(int, int, int);
void launch_wstream_gemm(bf16_t*, const bf16_t*, const bf16_t*, float*, int, int, int, int, hipStream_t);

namespace {

inline hipStream_t stream() {
  return at::cuda::getCurrentCUDAStream().stream();
}

inline bf16_t* bf(torch::Tensor& t) {
  return reinterpret_cast<bf16_t*>(t.data_ptr<at::BFloat16>());
}
inline const bf16_t* bfc(const torch::Tensor& t) {
  return reinterpret_cast<const bf16_t*>(t.data_ptr<at::BFloat16>());
}

// [ntok, h, d] tensors may be views of a fused qkv projection: inner two
// dims contiguous, token-row stride arbitrary (saves the decode path three
// elementwise copies per layer)
void check_bf16_rows(const torch::Tensor& t, const char* name) {
  TORCH_CHECK(t.is_cuda(), name, " must be on GPU");
  TORCH_CHECK(t.scalar_type() == torch::kBFloat16, name, " must be bf16");
  TORCH_CHECK(t.dim() == 3 && t.stride(2) == 1 &&
                  t.stride(1) == t.size(2),
              name, " must be [n, h, d] with contiguous rows");
}

void check_bf16(const torch::Tensor& t, const char* name) {
  TORCH_CHECK(t.is_cuda(), name, " must be on GPU");
  TORCH_CHECK(t.scalar_type() == torch::kBFloat16, name, " must be bf16");
  TORCH_CHECK(t.is_contiguous(), name, " must be contiguous");
}

void post_launch_check(const char* op) {
  hipError_t e = hipGetLastError();
  TORCH_CHECK(e == hipSuccess, op, ": HIP launch failed: ", hipGetErrorString(e));
}

void rms_norm(torch::Tensor out, torch::Tensor x, torch::Tensor w, double eps) {
  check_bf16(out, "out"); check_bf16(x, "x"); check_bf16(w, "w");
  int hidden = x.size(-1);
  TORCH_CHECK(hidden % 8 == 0, "hidden must be a multiple of 8");
  TORCH_CHECK(hidden <= 16384, "hidden > 16384 unsupported (register cache)");
  long rows = x.numel() / hidden;
  launch_rms_norm(bf(out), bfc(x), bfc(w), rows, hidden, (float)eps, stream());
  post_launch_check("rms_norm");
}

void rms_norm_residual(torch::Tensor out, torch::Tensor residual,
                       torch::Tensor x, torch::Tensor w, double eps) {
  check_bf16(out, "out"); check_bf16(residual, "residual");
  check_bf16(x, "x"); check_bf16(w, "w");
  int hidden = x.size(-1);
  TORCH_CHECK(hidden % 8 == 0, "hidden must be a multiple of 8");
  TORCH_CHECK(hidden <= 16384, "hidden > 16384 unsupported (register cache)");
  long rows = x.numel() / hidden;
  launch_rms_norm_residual(bf(out), bf(residual), bfc(x), bfc(w), rows,
                           hidden, (float)eps, stream());
  post_launch_check("rms_norm_residual");
}

void layer_norm(torch::Tensor out, torch::Tensor x, torch::Tensor w,
                torch::Tensor b, double eps) {
  check_bf16(out, "out"); check_bf16(x, "x"); check_bf16(w, "w");
  check_bf16(b, "b");
  int hidden = x.size(-1);
  TORCH_CHECK(hidden % 8 == 0, "hidden must be a multiple of 8");
  long rows = x.numel() / hidden;
  launch_layer_norm(bf(out), bfc(x), bfc(w), bfc(b), rows, hidden, (float)eps,
                    stream());
  post_launch_check("layer_norm");
}

void silu_mul(torch::Tensor out, torch::Tensor x) {
  check_bf16(out, "out"); check_bf16(x, "x");
  int f = out.size(-1);
  TORCH_CHECK(x.size(-1) == 2 * f, "x last dim must be 2*out last dim");
  TORCH_CHECK(f % 8 == 0, "intermediate size must be a multiple of 8");
  long rows = out.numel() / f;
  launch_silu_mul(bf(out), bfc(x), rows, f, stream());
  post_launch_check("silu_mul");
}

void gelu_tanh(torch::Tensor out, torch::Tensor x) {
  check_bf16(out, "out"); check_bf16(x, "x");
  TORCH_CHECK(x.numel() % 8 == 0, "numel must be a multiple of 8");
  launch_gelu(bf(out), bfc(x), x.numel(), stream());
  post_launch_check("gelu_tanh");
}

bool cache_kv8(const torch::Tensor& t, const char* name) {
  if (t.scalar_type() == torch::kBFloat16) return false;
  TORCH_CHECK(t.scalar_type() == torch::kFloat8_e4m3fn, name,
              " cache must be bf16 or float8_e4m3fn");
  return true;
}

void rope_kv_append(torch::Tensor q, torch::Tensor k, torch::Tensor v,
                    torch::Tensor positions, torch::Tensor k_cache,
                    torch::Tensor v_cache, torch::Tensor slots, double theta) {
  check_bf16_rows(q, "q"); check_bf16_rows(k, "k"); check_bf16_rows(v, "v");
  const bool kv8 = cache_kv8(k_cache, "k");
  TORCH_CHECK(cache_kv8(v_cache, "v") == kv8, "cache dtypes must match");
  TORCH_CHECK(k_cache.is_contiguous() && v_cache.is_contiguous());
  TORCH_CHECK(positions.scalar_type() == torch::kInt, "positions must be int32");
  TORCH_CHECK(slots.scalar_type() == torch::kInt, "slots must be int32");
  int ntok = q.size(0), hq = q.size(1), d = q.size(2);
  int hkv = k.size(1);
  int block_size = k_cache.size(2);
  TORCH_CHECK(k_cache.size(1) == hkv && k_cache.size(3) == d, "cache shape");
  TORCH_CHECK(d % 8 == 0 && (d / 2) % 8 == 0, "head_dim constraints");
  launch_rope_kv_append(bf(q), bf(k), bfc(v), positions.data_ptr<int>(),
                        k_cache.data_ptr(), v_cache.data_ptr(),
                        slots.data_ptr<int>(), ntok,
                        hq, hkv, d, block_size, (float)theta, kv8,
                        q.stride(0), k.stride(0), v.stride(0), stream());
  post_launch_check("rope_kv_append");
}

void paged_attn_decode(torch::Tensor out, torch::Tensor q,
                       torch::Tensor k_cache, torch::Tensor v_cache,
                       torch::Tensor block_table, torch::Tensor seq_lens,
                       double scale) {
  check_bf16(out, "out"); check_bf16_rows(q, "q");
  const bool kv8 = cache_kv8(k_cache, "k");
  TORCH_CHECK(cache_kv8(v_cache, "v") == kv8, "cache dtypes must match");
  TORCH_CHECK(k_cache.is_contiguous() && v_cache.is_contiguous());
  TORCH_CHECK(block_table.scalar_type() == torch::kInt, "block_table int32");
  TORCH_CHECK(seq_lens.scalar_type() == torch::kInt, "seq_lens int32");
  TORCH_CHECK(block_table.is_contiguous() && seq_lens.is_contiguous());
  int bsz = q.size(0), hq = q.size(1), d = q.size(2);
  int hkv = k_cache.size(1), block_size = k_cache.size(2);
  int max_blocks = block_table.size(1);
  TORCH_CHECK(paged_attn_decode_supported(d, block_size),
              "paged_attn_decode: unsupported head_dim ", d, " / block_size ",
              block_size);
  TORCH_CHECK(hq % hkv == 0, "hq must be divisible by hkv");
  const int nsplit = paged_attn_nsplit(bsz, hq, hkv, d);
  float *po = nullptr, *pml = nullptr;
  torch::Tensor part_o, part_ml;
  if (nsplit > 1) {
    auto opts = q.options().dtype(torch::kFloat);
    part_o = torch::empty({bsz, hq, nsplit, d}, opts);
    part_ml = torch::empty({bsz, hq, nsplit, 2}, opts);
    po = part_o.data_ptr<float>();
    pml = part_ml.data_ptr<float>();
  }
  launch_paged_attn_decode(bf(out), bfc(q), k_cache.data_ptr(),
                           v_cache.data_ptr(),
                           block_table.data_ptr<int>(),
                           seq_lens.data_ptr<int>(), bsz, hq, hkv, d,
                           max_blocks, block_size, (float)scale, q.stride(0),
                           nsplit, po, pml, kv8, stream());
  post_launch_check("paged_attn_decode");
}

void attn_prefill(torch::Tensor out, torch::Tensor q, torch::Tensor k,
                  torch::Tensor v, torch::Tensor seq_start, long max_len,
                  double scale) {
  check_bf16(out, "out"); check_bf16(q, "q"); check_bf16(k, "k");
  check_bf16(v, "v");
  TORCH_CHECK(seq_start.scalar_type() == torch::kInt, "seq_start int32");
  int hq = q.size(1), d = q.size(2), hkv = k.size(1);
  int num_seqs = seq_start.size(0) - 1;
  TORCH_CHECK(attn_prefill_supported(d), "unsupported head_dim ", d);
  TORCH_CHECK(hq % hkv == 0, "hq must be divisible by hkv");
  launch_attn_prefill(bf(out), bfc(q), bfc(k), bfc(v),
                      seq_start.data_ptr<int>(), num_seqs, (int)max_len, hq,
                      hkv, d, (float)scale, stream(), nullptr);
  post_launch_check("attn_prefill");
}

void attn_v5(torch::Tensor out, torch::Tensor q, torch::Tensor k,
             torch::Tensor v, torch::Tensor seq_start, long max_len,
             double scale, c10::optional<torch::Tensor> lse) {
  check_bf16(out, "out"); check_bf16(q, "q"); check_bf16(k, "k");
  check_bf16(v, "v");
  TORCH_CHECK(seq_start.scalar_type() == torch::kInt, "seq_start int32");
  int hq = q.size(1), d = q.size(2), hkv = k.size(1);
  int num_seqs = seq_start.size(0) - 1;
  TORCH_CHECK(attn_v5_supported(d), "unsupported head_dim ", d);
  TORCH_CHECK(hq % hkv == 0, "hq must be divisible by hkv");
  float* lp = nullptr;
  if (lse.has_value()) {
    TORCH_CHECK(lse->scalar_type() == torch::kFloat, "lse f32");
    lp = lse->data_ptr<float>();
  }
  launch_attn_v5(bf(out), bfc(q), bfc(k), bfc(v),
                 seq_start.data_ptr<int>(), num_seqs, (int)max_len, hq,
                 hkv, d, (float)scale, stream(), lp);
  post_launch_check("attn_v5");
}

// forward with saved log2-domain LSE (training path)
void attn_train_fwd(torch::Tensor out, torch::Tensor lse, torch::Tensor q,
                    torch::Tensor k, torch::Tensor v,
                    torch::Tensor seq_start, long max_len, double scale) {
  check_bf16(out, "out"); check_bf16(q, "q"); check_bf16(k, "k");
  check_bf16(v, "v");
  TORCH_CHECK(lse.scalar_type() == torch::kFloat, "lse must be f32");
  TORCH_CHECK(seq_start.scalar_type() == torch::kInt, "seq_start int32");
  int hq = q.size(1), d = q.size(2), hkv = k.size(1);
  int num_seqs = seq_start.size(0) - 1;
  TORCH_CHECK(attn_prefill_supported(d), "unsupported head_dim ", d);
  TORCH_CHECK(hq % hkv == 0, "hq must be divisible by hkv");
  TORCH_CHECK(lse.numel() == q.size(0) * hq, "lse shape");
  launch_attn_prefill(bf(out), bfc(q), bfc(k), bfc(v),
                      seq_start.data_ptr<int>(), num_seqs, (int)max_len, hq,
                      hkv, d, (float)scale, stream(),
                      lse.data_ptr<float>());
  post_launch_check("attn_train_fwd");
}

void attn_train_bwd(torch::Tensor dq, torch::Tensor dk, torch::Tensor dv,
                    torch::Tensor q, torch::Tensor k, torch::Tensor v,
                    torch::Tensor out, torch::Tensor dout,
                    torch::Tensor lse, torch::Tensor delta,
                    torch::Tensor seq_start, long max_len, double scale) {
  check_bf16(dq, "dq");
  TORCH_CHECK(dk.scalar_type() == torch::kFloat &&
              dv.scalar_type() == torch::kFloat,
              "dk/dv are f32 accumulators (GQA heads atomically add)");
  check_bf16(q, "q"); check_bf16(k, "k"); check_bf16(v, "v");
  check_bf16(out, "out"); check_bf16(dout, "dout");
  TORCH_CHECK(lse.scalar_type() == torch::kFloat &&
              delta.scalar_type() == torch::kFloat, "lse/delta f32");
  TORCH_CHECK(seq_start.scalar_type() == torch::kInt, "seq_start int32");
  int hq = q.size(1), d = q.size(2), hkv = k.size(1);
  int num_seqs = seq_start.size(0) - 1;
  TORCH_CHECK(attn_bwd_supported(d), "unsupported head_dim ", d);
  TORCH_CHECK(hq % hkv == 0, "hq must be divisible by hkv");
  launch_attn_bwd_delta(delta.data_ptr<float>(), bfc(dout), bfc(out),
                        (long)q.size(0) * hq, d, stream());
  launch_attn_bwd(bf(dq), dk.data_ptr<float>(), dv.data_ptr<float>(),
                  bfc(q), bfc(k), bfc(v), bfc(dout),
                  lse.data_ptr<float>(), delta.data_ptr<float>(),
                  seq_start.data_ptr<int>(), num_seqs, (int)max_len, hq,
                  hkv, d, (float)scale, stream());
  post_launch_check("attn_train_bwd");
}
void attn_train_bwd_v5(torch::Tensor dq, torch::Tensor dk,
                       torch::Tensor dv, torch::Tensor q, torch::Tensor k,
                       torch::Tensor v, torch::Tensor out,
                       torch::Tensor dout, torch::Tensor lse,
                       torch::Tensor delta, torch::Tensor seq_start,
                       long max_len, double scale) {
  check_bf16(dq, "dq");
  TORCH_CHECK(dk.scalar_type() == torch::kFloat &&
              dv.scalar_type() == torch::kFloat, "dk/dv f32");
  check_bf16(q, "q"); check_bf16(k, "k"); check_bf16(v, "v");
  check_bf16(out, "out"); check_bf16(dout, "dout");
  TORCH_CHECK(lse.scalar_type() == torch::kFloat &&
              delta.scalar_type() == torch::kFloat, "lse/delta f32");
  TORCH_CHECK(seq_start.scalar_type() == torch::kInt, "seq_start int32");
  int hq = q.size(1), d = q.size(2), hkv = k.size(1);
  int num_seqs = seq_start.size(0) - 1;
  TORCH_CHECK(attn_v5_supported(d), "unsupported head_dim ", d);
  TORCH_CHECK(hq % hkv == 0, "hq must be divisible by hkv");
  launch_attn_bwd_delta(delta.data_ptr<float>(), bfc(dout), bfc(out),
                        (long)q.size(0) * hq, d, stream());
  launch_attn_bwd_dq_v5(bf(dq), bfc(q), bfc(k), bfc(v), bfc(dout),
                        lse.data_ptr<float>(), delta.data_ptr<float>(),
                        seq_start.data_ptr<int>(), num_seqs, (int)max_len,
                        hq, hkv, d, (float)scale, stream());
  // dkv dispatch by GQA group size: the V5-structure dkv WINS at grp==1
  // (llama MHA: 1010 vs 1113 us) but loses 2x under MQA/GQA — its fewer,
  // wider blocks collide harder on the per-kv-row atomic adds (falcon
  // 71:1: 2477 vs 918 us). SUB_ATTN_DKV_V5=0/1 overrides.
  const char* e = getenv("SUB_ATTN_DKV_V5");
  const bool use_v5 = e ? (e[0] == '1') : (hq == hkv);
  if (use_v5)
    launch_attn_bwd_dkv_v5(dk.data_ptr<float>(), dv.data_ptr<float>(),
                           bfc(q), bfc(k), bfc(v), bfc(dout),
                           lse.data_ptr<float>(), delta.data_ptr<float>(),
                           seq_start.data_ptr<int>(), num_seqs,
                           (int)max_len, hq, hkv, d, (float)scale,
                           stream());
  else
    launch_attn_bwd_dkv_only(dk.data_ptr<float>(), dv.data_ptr<float>(),
                             bfc(q), bfc(k), bfc(v), bfc(dout),
                             lse.data_ptr<float>(), delta.data_ptr<float>(),
                             seq_start.data_ptr<int>(), num_seqs,
                             (int)max_len, hq, hkv, d, (float)scale,
                             stream());
  post_launch_check("attn_train_bwd_v5");
}


void attn_prefill_probe(long variant, torch::Tensor out, torch::Tensor q,
                        torch::Tensor k, torch::Tensor v,
                        torch::Tensor seq_start, long max_len, double scale) {
  check_bf16(out, "out"); check_bf16(q, "q"); check_bf16(k, "k");
  check_bf16(v, "v");
  int hq = q.size(1), hkv = k.size(1);
  TORCH_CHECK(q.size(2) == 128, "probe is D=128 only");
  launch_attn_prefill_probe((int)variant, bf(out), bfc(q), bfc(k), bfc(v),
                            seq_start.data_ptr<int>(),
                            seq_start.size(0) - 1, (int)max_len, hq, hkv,
                            (float)scale, stream());
  post_launch_check("attn_prefill_probe");
}

void mfma_gemm(torch::Tensor y, torch::Tensor x, torch::Tensor w,
               c10::optional<torch::Tensor> bias,
               c10::optional<torch::Tensor> part, long splitk) {
  check_bf16(y, "y"); check_bf16(x, "x"); check_bf16(w, "w");
  int m = x.size(0), k = x.size(1), n = w.size(0);
  TORCH_CHECK(w.size(1) == k, "shape mismatch");
  TORCH_CHECK(mfma_gemm_supported(m, n, k), "unsupported mfma_gemm shape ",
              m, "x", n, "x", k);
  const bf16_t* bp = nullptr;
  if (bias.has_value()) { check_bf16(*bias, "bias"); bp = bfc(*bias); }
  float* pp = nullptr;
  if (splitk > 1) {
    TORCH_CHECK(part.has_value() &&
                part->numel() >= splitk * (long)m * n, "part too small");
    pp = part->data_ptr<float>();
  }
  launch_mfma_gemm(bf(y), bfc(x), bfc(w), bp, pp, m, n, k, (int)splitk,
                   stream());
  post_launch_check("mfma_gemm");
}

long mfma_gemm_pick_splitk(long m, long n, long k) {
  if (!mfma_gemm_supported((int)m, (int)n, (int)k)) return 0;
  return mfma_gemm_splitk((int)m, (int)n, (int)k);
}

void quant_gemv_k(torch::Tensor y, torch::Tensor x, torch::Tensor scales,
                  torch::Tensor qs, bool q6) {
  check_bf16(y, "y"); check_bf16(x, "x");
  TORCH_CHECK(scales.scalar_type() == torch::kHalf, "scales must be fp16");
  TORCH_CHECK(qs.scalar_type() == torch::kByte, "qs must be uint8");
  int m = x.size(0), k = x.size(1), n = y.size(1);
  TORCH_CHECK(m >= 1 && m <= 8, "quant_gemv_k is the M<=8 decode path");
  TORCH_CHECK(k % 256 == 0, "K must be a multiple of 256 (K-quants)");
  launch_quant_gemv_k(bf(y), bfc(x), scales.data_ptr(),
                      qs.data_ptr<unsigned char>(), m, n, k, q6, stream());
  post_launch_check("quant_gemv_k");
}

void dequant_k(torch::Tensor out, torch::Tensor scales, torch::Tensor qs,
               bool q6) {
  check_bf16(out, "out");
  TORCH_CHECK(scales.scalar_type() == torch::kHalf, "scales must be fp16");
  TORCH_CHECK(qs.scalar_type() == torch::kByte, "qs must be uint8");
  int n = out.size(0), k = out.size(1);
  TORCH_CHECK(k % 256 == 0, "K must be a multiple of 256 (K-quants)");
  launch_dequant_k(bf(out), scales.data_ptr(),
                   qs.data_ptr<unsigned char>(), n, k, q6, stream());
  post_launch_check("dequant_k");
}

torch::Tensor mfma32_probe(torch::Tensor a, torch::Tensor b) {
  check_bf16(a, "a"); check_bf16(b, "b");
  TORCH_CHECK(a.size(0) == 32 && a.size(1) == 16 &&
              b.size(0) == 16 && b.size(1) == 32, "probe is 32x16 x 16x32");
  auto d = torch::empty({32, 32}, a.options().dtype(torch::kFloat));
  launch_mfma32_probe(d.data_ptr<float>(), bfc(a), bfc(b), stream());
  post_launch_check("mfma32_probe");
  return d;
}

torch::Tensor mfma_probe(torch::Tensor a, torch::Tensor b) {
  check_bf16(a, "a"); check_bf16(b, "b");
  TORCH_CHECK(a.size(0) == 16 && a.size(1) == 32 && b.size(0) == 32 &&
              b.size(1) == 16, "probe shapes: a[16,32] b[32,16]");
  auto out = torch::empty({16, 16}, a.options().dtype(torch::kFloat));
  launch_mfma_probe(out.data_ptr<float>(), bfc(a), bfc(b), stream());
  post_launch_check("mfma_probe");
  return out;
}

void softmax_sample(torch::Tensor out, torch::Tensor logits,
                    torch::Tensor temperature, torch::Tensor top_k,
                    torch::Tensor top_p, torch::Tensor uniform) {
  check_bf16(logits, "logits");
  TORCH_CHECK(out.scalar_type() == torch::kLong, "out must be int64");
  int bsz = logits.size(0), vocab = logits.size(1);
  TORCH_CHECK((size_t)vocab * 2 <= 160 * 1024 - 4096,
              "vocab too large for LDS staging: ", vocab);
  launch_softmax_sample(out.data_ptr<long>(), bfc(logits),
                        temperature.data_ptr<float>(), top_k.data_ptr<int>(),
                        top_p.data_ptr<float>(), uniform.data_ptr<float>(),
                        bsz, vocab, stream());
  post_launch_check("softmax_sample");
}

void cross_entropy_fwd_bwd(torch::Tensor loss, torch::Tensor dlogits,
                           torch::Tensor logits, torch::Tensor targets,
                           long ignore_index) {
  check_bf16(dlogits, "dlogits"); check_bf16(logits, "logits");
  TORCH_CHECK(loss.scalar_type() == torch::kFloat, "loss must be fp32");
  TORCH_CHECK(targets.scalar_type() == torch::kLong, "targets must be int64");
  long rows = logits.size(0);
  int vocab = logits.size(1);
  TORCH_CHECK(vocab % 8 == 0, "vocab must be a multiple of 8");
  long nvalid = (targets != ignore_index).sum().item<long>();
  float inv = nvalid > 0 ? 1.f / (float)nvalid : 0.f;
  launch_cross_entropy(loss.data_ptr<float>(), bf(dlogits), bfc(logits),
                       targets.data_ptr<long>(), rows, vocab, ignore_index,
                       inv, stream());
  post_launch_check("cross_entropy_fwd_bwd");
}

void adamw_step(torch::Tensor p, torch::Tensor g, torch::Tensor m,
                torch::Tensor v, double lr, double beta1, double beta2,
                double eps, double wd, long step, double gscale) {
  TORCH_CHECK(p.is_cuda() && p.is_contiguous());
  TORCH_CHECK(m.scalar_type() == torch::kFloat && v.scalar_type() == torch::kFloat);
  long n = p.numel();
  TORCH_CHECK(g.numel() == n && m.numel() == n && v.numel() == n);
  if (p.scalar_type() == torch::kBFloat16) {
    TORCH_CHECK(g.scalar_type() == torch::kBFloat16);
    launch_adamw(bf(p), bfc(g), m.data_ptr<float>(), v.data_ptr<float>(), n,
                 (float)lr, (float)beta1, (float)beta2, (float)eps, (float)wd,
                 (int)step, (float)gscale, stream());
  } else {
    TORCH_CHECK(p.scalar_type() == torch::kFloat && g.scalar_type() == torch::kFloat);
    launch_adamw_f32(p.data_ptr<float>(), g.data_ptr<float>(),
                     m.data_ptr<float>(), v.data_ptr<float>(), n, (float)lr,
                     (float)beta1, (float)beta2, (float)eps, (float)wd,
                     (int)step, (float)gscale, stream());
  }
  post_launch_check("adamw_step");
}

void quant_gemv(torch::Tensor y, torch::Tensor x, torch::Tensor scales,
                torch::Tensor qs, bool q8) {
  check_bf16(y, "y"); check_bf16(x, "x");
  TORCH_CHECK(scales.scalar_type() == torch::kHalf, "scales must be fp16");
  TORCH_CHECK(qs.scalar_type() == torch::kByte, "qs must be uint8");
  TORCH_CHECK(scales.is_contiguous() && qs.is_contiguous());
  int m = x.size(0), k = x.size(1), n = y.size(1);
  TORCH_CHECK(m >= 1 && m <= 8, "quant_gemv supports M in [1,8], got ", m);
  TORCH_CHECK(k % 32 == 0, "K must be a multiple of 32");
  launch_quant_gemv(bf(y), bfc(x), scales.data_ptr(),
                    qs.data_ptr<unsigned char>(), m, n, k, q8, stream());
  post_launch_check("quant_gemv");
}

void dequant(torch::Tensor out, torch::Tensor scales, torch::Tensor qs,
             bool q8) {
  check_bf16(out, "out");
  TORCH_CHECK(scales.scalar_type() == torch::kHalf && qs.scalar_type() == torch::kByte);
  long n = out.size(0);
  int k = out.size(1);
  launch_dequant(bf(out), scales.data_ptr(), qs.data_ptr<unsigned char>(),
                 n, k, q8, stream());
  post_launch_check("dequant");
}

void decode_gemm(torch::Tensor y, torch::Tensor x, torch::Tensor w) {
  check_bf16(y, "y"); check_bf16(x, "x"); check_bf16(w, "w");
  int m = x.size(0), k = x.size(1), n = w.size(0);
  TORCH_CHECK(w.size(1) == k && y.size(0) == m && y.size(1) == n);
  TORCH_CHECK(decode_gemm_supported(m, n, k),
              "decode_gemm unsupported shape M=", m, " N=", n, " K=", k);
  const int splitk = decode_gemm_splitk(n, k);
  auto part = torch::empty({splitk, 256, n},
                           x.options().dtype(torch::kFloat));
  launch_decode_gemm(bf(y), bfc(x), bfc(w), part.data_ptr<float>(), m, n, k,
                     splitk, stream());
  post_launch_check("decode_gemm");
}

bool decode_gemm_ok(long m, long n, long k) {
  return decode_gemm_supported((int)m, (int)n, (int)k);
}

// Y[M, N] = X[M, K] · W^T with W repacked fragment-linear
// (ops.wstream_repack); wstream_gemm.hip
void wstream_gemm(torch::Tensor y, torch::Tensor x, torch::Tensor wp,
                  long n_, long k_) {
  check_bf16(y, "y"); check_bf16(x, "x"); check_bf16(wp, "wp");
  const int m = x.size(0), k = (int)k_, n = (int)n_;
  TORCH_CHECK(x.size(1) == k && y.size(0) == m && y.size(1) == n);
  TORCH_CHECK(wp.numel() == (long)n * k, "packed W size mismatch");
  TORCH_CHECK(wstream_supported(m, n, k),
              "wstream_gemm unsupported shape M=", m, " N=", n, " K=", k);
  const int splitk = wstream_splitk(n, k);
  torch::Tensor part;
  float* pp = nullptr;
  if (splitk > 1) {
    part = torch::empty({splitk, 256, n}, x.options().dtype(torch::kFloat));
    pp = part.data_ptr<float>();
  }
  launch_wstream_gemm(bf(y), bfc(x), bfc(wp), pp, m, n, k, splitk, stream());
  post_launch_check("wstream_gemm");
}

bool wstream_ok(long m, long n, long k) {
  return wstream_supported((int)m, (int)n, (int)k);
}

}  // namespace

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
  m.def("decode_gemm", &decode_gemm);
  m.def("decode_gemm_ok", &decode_gemm_ok);
  m.def("wstream_gemm", &wstream_gemm);
  m.def("wstream_ok", &wstream_ok);
  m.def("quant_gemv", &quant_gemv);
  m.def("dequant", &dequant);
  m.def("rms_norm", &rms_norm);
  m.def("rms_norm_residual", &rms_norm_residual);
  m.def("layer_norm", &layer_norm);
  m.def("silu_mul", &silu_mul);
  m.def("gelu_tanh", &gelu_tanh);
  m.def("rope_kv_append", &rope_kv_append);
  m.def("paged_attn_decode", &paged_attn_decode);
  m.def("attn_prefill", &attn_prefill);
  m.def("attn_train_fwd", &attn_train_fwd);
  m.def("mfma_gemm", &mfma_gemm);
  m.def("quant_gemv_k", &quant_gemv_k);
  m.def("mfma32_probe", &mfma32_probe);
  m.def("attn_v5", &attn_v5);
  m.def("attn_train_bwd_v5", &attn_train_bwd_v5);
  m.def("dequant_k", &dequant_k);
  m.def("mfma_gemm_pick_splitk", &mfma_gemm_pick_splitk);
  m.def("attn_train_bwd", &attn_train_bwd);
  m.def("mfma_probe", &mfma_probe);
  m.def("attn_prefill_probe", &attn_prefill_probe);
  m.def("softmax_sample", &softmax_sample);
  m.def("cross_entropy_fwd_bwd", &cross_entropy_fwd_bwd);
  m.def("adamw_step", &adamw_step);
}
