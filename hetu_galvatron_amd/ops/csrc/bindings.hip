// Torch bindings for the gfx950 kernel suite (module: _galvatron_hip).
//
// HIP-native throughout (c10::hip stream API, no CUDA-compat shims).
// Python-side dispatch contract: hetu_galvatron_amd/ops/functional.py,
// runtime/tensor_parallel/cross_entropy.py, runtime/optimizer/optimizer.py.
#include <torch/extension.h>
#include <c10/hip/HIPStream.h>
#include <c10/hip/HIPGuard.h>

#include <hip/hip_runtime.h>

#include <tuple>
#include <vector>

// ---- launcher prototypes (defined in the sibling .hip TUs) ----------------
template <typename T>
void rmsnorm_fwd_launch_t(const T*, const T*, T*, float*, long, int, float, hipStream_t, const T* res = nullptr, T* sum_out = nullptr);
template <typename T>
void rmsnorm_bwd_launch_t(const T*, const T*, const T*, const float*, T*, float*, float*, long, int, hipStream_t, const T* dsum_in = nullptr);
template <typename T>
void layernorm_fwd_launch_t(const T*, const T*, const T*, T*, float*, float*, long, int, float, hipStream_t);
template <typename T>
void layernorm_bwd_launch_t(const T*, const T*, const T*, const float*, const float*, T*, float*, float*, long, int, hipStream_t);
int norm_bwd_grid(long n);
template <typename T>
void swiglu_fwd_launch_t(const T*, T*, long, int, hipStream_t);
template <typename T>
void swiglu_bwd_launch_t(const T*, const T*, T*, long, int, hipStream_t);
template <typename T>
void rope_launch_t(const T*, T*, const float*, const float*, long, int, int, bool, hipStream_t);
template <typename TG>
void grad_accum_launch_t(float*, const TG*, long, hipStream_t);
template <typename TG, typename TO>
void adamw_launch_t(float*, const TG*, float*, float*, TO*, long, int, float, float, float, float, float, float, hipStream_t);
template <typename T>
void ce_max_launch_t(const T*, float*, long, long, hipStream_t);
template <typename T>
void ce_sum_target_launch_t(const T*, const long*, const float*, float*, float*, long, long, long, hipStream_t);
template <typename T>
void ce_bwd_launch_t(T*, const long*, const float*, const float*, const float*, long, long, long, hipStream_t);

void grouped_gemm_launch(const __bf16*, const __bf16*, __bf16*, const void*, int, int, int, long, bool, hipStream_t);
void grouped_gemm_dw_launch(const __bf16*, const __bf16*, float*, const int*, int, int, int, hipStream_t);
#define DECL_MOE(T) \
  void moe_permute_launch_##T(const T*, T*, const long*, long, int, hipStream_t); \
  void moe_permute_bwd_launch_##T(const T*, float*, const long*, long, int, hipStream_t); \
  void moe_unpermute_launch_##T(const T*, const float*, const long*, T*, long, int, int, hipStream_t); \
  void moe_unpermute_bwd_launch_##T(const T*, const T*, const float*, const long*, T*, float*, long, int, hipStream_t);
typedef __bf16 bf16_t;
typedef float f32_t;
DECL_MOE(bf16_t)
DECL_MOE(f32_t)

void flash_fwd_launch(const __bf16*, const __bf16*, const __bf16*, __bf16*, float*, int, int, int, int, int, int, float, bool, hipStream_t, const __bf16* bias = nullptr, bool sbhd = false, int window = 0);
void attn_di_launch(const __bf16*, const __bf16*, float*, int, int, int, int, hipStream_t, bool sbhd = false);
void flash_bwd_launch(const __bf16*, const __bf16*, const __bf16*, const __bf16*, const float*, const float*, __bf16*, __bf16*, __bf16*, int, int, int, int, int, int, float, bool, hipStream_t, const __bf16* bias = nullptr, float* dbias = nullptr, bool sbhd = false, int window = 0);
void multi_sumsq_launch(const long*, int, float*, hipStream_t);
void mfma_probe_launch(const __bf16*, const __bf16*, float*, bool, hipStream_t);
void decode_attn_launch(const __bf16*, const __bf16*, const __bf16*, __bf16*, float*, int, int, int, int, int, int, int, float, const int*, hipStream_t);

namespace {

using at::Tensor;

hipStream_t cur_stream() { return c10::hip::getCurrentHIPStream().stream(); }

#define CHECK_GPU(x) \
  TORCH_CHECK(x.is_cuda() && x.is_contiguous(), #x " must be contiguous on GPU")

const __bf16* bfp(const Tensor& t) {
  return reinterpret_cast<const __bf16*>(t.data_ptr<at::BFloat16>());
}
__bf16* bfp_mut(Tensor& t) {
  return reinterpret_cast<__bf16*>(t.data_ptr<at::BFloat16>());
}

bool is_bf16(const Tensor& t) { return t.scalar_type() == at::kBFloat16; }

// ---- norms ----------------------------------------------------------------
std::tuple<Tensor, Tensor> rmsnorm_fwd(const Tensor& x, const Tensor& w,
                                       double eps) {
  CHECK_GPU(x);
  const int H = x.size(-1);
  const long n = x.numel() / H;
  TORCH_CHECK(H % 8 == 0 && H <= 16384, "rmsnorm: H must be %8, <=16k");
  auto y = at::empty_like(x);
  auto invrms = at::empty({n}, x.options().dtype(at::kFloat));
  if (is_bf16(x))
    rmsnorm_fwd_launch_t<__bf16>(bfp(x), bfp(w), bfp_mut(y),
                                 invrms.data_ptr<float>(), n, H, (float)eps,
                                 cur_stream());
  else
    rmsnorm_fwd_launch_t<float>(x.data_ptr<float>(), w.data_ptr<float>(),
                                y.data_ptr<float>(), invrms.data_ptr<float>(),
                                n, H, (float)eps, cur_stream());
  auto sizes = x.sizes().vec();
  sizes.pop_back();
  return {y, invrms.view(sizes)};
}

std::tuple<Tensor, Tensor> rmsnorm_bwd(const Tensor& dy, const Tensor& x,
                                       const Tensor& w, const Tensor& invrms) {
  CHECK_GPU(dy);
  const int H = x.size(-1);
  const long n = x.numel() / H;
  auto dx = at::empty_like(x);
  auto dw = at::zeros({H}, x.options().dtype(at::kFloat));
  const int G = norm_bwd_grid(n);
  auto dw_part = at::empty({G, H}, x.options().dtype(at::kFloat));
  auto inv = invrms.contiguous();
  if (is_bf16(x))
    rmsnorm_bwd_launch_t<__bf16>(bfp(dy), bfp(x), bfp(w),
                                 inv.data_ptr<float>(), bfp_mut(dx),
                                 dw_part.data_ptr<float>(),
                                 dw.data_ptr<float>(), n, H, cur_stream());
  else
    rmsnorm_bwd_launch_t<float>(dy.data_ptr<float>(), x.data_ptr<float>(),
                                w.data_ptr<float>(), inv.data_ptr<float>(),
                                dx.data_ptr<float>(), dw_part.data_ptr<float>(),
                                dw.data_ptr<float>(), n, H, cur_stream());
  return {dx, dw};
}

// fused residual-add + rmsnorm: y = rmsnorm(x + res) * w, also returns
// the sum (the downstream residual stream) and invrms
std::tuple<Tensor, Tensor, Tensor> add_rmsnorm_fwd(const Tensor& x,
                                                   const Tensor& res,
                                                   const Tensor& w,
                                                   double eps) {
  CHECK_GPU(x); CHECK_GPU(res);
  TORCH_CHECK(is_bf16(x) && is_bf16(res), "add_rmsnorm: bf16 only");
  const int H = x.size(-1);
  const long n = x.numel() / H;
  TORCH_CHECK(H % 8 == 0 && H <= 16384, "rmsnorm: H must be %8, <=16k");
  auto y = at::empty_like(x);
  auto sum = at::empty_like(x);
  auto invrms = at::empty({n}, x.options().dtype(at::kFloat));
  rmsnorm_fwd_launch_t<__bf16>(bfp(x), bfp(w), bfp_mut(y),
                               invrms.data_ptr<float>(), n, H, (float)eps,
                               cur_stream(), bfp(res), bfp_mut(sum));
  auto sizes = x.sizes().vec();
  sizes.pop_back();
  return {y, sum, invrms.view(sizes)};
}

// backward with the residual-stream grad folded in:
// dx = rmsnorm_bwd(dy wrt sum) + dsum  (dx == dres)
std::tuple<Tensor, Tensor> add_rmsnorm_bwd(const Tensor& dy,
                                           const Tensor& dsum,
                                           const Tensor& sum,
                                           const Tensor& w,
                                           const Tensor& invrms) {
  CHECK_GPU(dy);
  const int H = sum.size(-1);
  const long n = sum.numel() / H;
  auto dx = at::empty_like(sum);
  auto dw = at::zeros({H}, sum.options().dtype(at::kFloat));
  const int G = norm_bwd_grid(n);
  auto dw_part = at::empty({G, H}, sum.options().dtype(at::kFloat));
  auto inv = invrms.contiguous();
  rmsnorm_bwd_launch_t<__bf16>(bfp(dy), bfp(sum), bfp(w),
                               inv.data_ptr<float>(), bfp_mut(dx),
                               dw_part.data_ptr<float>(),
                               dw.data_ptr<float>(), n, H, cur_stream(),
                               bfp(dsum));
  return {dx, dw};
}

std::tuple<Tensor, Tensor, Tensor> layernorm_fwd(const Tensor& x,
                                                 const Tensor& w,
                                                 const Tensor& b, double eps) {
  CHECK_GPU(x);
  const int H = x.size(-1);
  const long n = x.numel() / H;
  TORCH_CHECK(H % 8 == 0 && H <= 16384, "layernorm: H must be %8, <=16k");
  auto y = at::empty_like(x);
  auto mean = at::empty({n}, x.options().dtype(at::kFloat));
  auto invstd = at::empty({n}, x.options().dtype(at::kFloat));
  if (is_bf16(x))
    layernorm_fwd_launch_t<__bf16>(bfp(x), bfp(w), bfp(b), bfp_mut(y),
                                   mean.data_ptr<float>(),
                                   invstd.data_ptr<float>(), n, H, (float)eps,
                                   cur_stream());
  else
    layernorm_fwd_launch_t<float>(x.data_ptr<float>(), w.data_ptr<float>(),
                                  b.data_ptr<float>(), y.data_ptr<float>(),
                                  mean.data_ptr<float>(), invstd.data_ptr<float>(),
                                  n, H, (float)eps, cur_stream());
  auto sizes = x.sizes().vec();
  sizes.pop_back();
  return {y, mean.view(sizes), invstd.view(sizes)};
}

std::tuple<Tensor, Tensor, Tensor> layernorm_bwd(const Tensor& dy,
                                                 const Tensor& x,
                                                 const Tensor& w,
                                                 const Tensor& mean,
                                                 const Tensor& invstd) {
  CHECK_GPU(dy);
  const int H = x.size(-1);
  const long n = x.numel() / H;
  auto dx = at::empty_like(x);
  auto dwdb = at::zeros({2, H}, x.options().dtype(at::kFloat));
  const int G = norm_bwd_grid(n);
  auto part = at::empty({2, G, H}, x.options().dtype(at::kFloat));
  auto m = mean.contiguous(), r = invstd.contiguous();
  if (is_bf16(x))
    layernorm_bwd_launch_t<__bf16>(bfp(dy), bfp(x), bfp(w),
                                   m.data_ptr<float>(), r.data_ptr<float>(),
                                   bfp_mut(dx), part.data_ptr<float>(),
                                   dwdb.data_ptr<float>(), n, H, cur_stream());
  else
    layernorm_bwd_launch_t<float>(dy.data_ptr<float>(), x.data_ptr<float>(),
                                  w.data_ptr<float>(), m.data_ptr<float>(),
                                  r.data_ptr<float>(), dx.data_ptr<float>(),
                                  part.data_ptr<float>(),
                                  dwdb.data_ptr<float>(), n, H, cur_stream());
  return {dx, dwdb[0], dwdb[1]};
}

// ---- swiglu / rope --------------------------------------------------------
Tensor swiglu_fwd(const Tensor& x) {
  CHECK_GPU(x);
  const int F2 = x.size(-1);
  TORCH_CHECK(F2 % 16 == 0, "swiglu: last dim must be %16");
  const int F = F2 / 2;
  const long rows = x.numel() / F2;
  auto sizes = x.sizes().vec();
  sizes.back() = F;
  auto y = at::empty(sizes, x.options());
  if (is_bf16(x))
    swiglu_fwd_launch_t<__bf16>(bfp(x), bfp_mut(y), rows, F, cur_stream());
  else
    swiglu_fwd_launch_t<float>(x.data_ptr<float>(), y.data_ptr<float>(),
                               rows, F, cur_stream());
  return y;
}

Tensor swiglu_bwd(const Tensor& dy, const Tensor& x) {
  CHECK_GPU(dy);
  const int F2 = x.size(-1);
  const int F = F2 / 2;
  const long rows = x.numel() / F2;
  auto dx = at::empty_like(x);
  if (is_bf16(x))
    swiglu_bwd_launch_t<__bf16>(bfp(dy), bfp(x), bfp_mut(dx), rows, F,
                                cur_stream());
  else
    swiglu_bwd_launch_t<float>(dy.data_ptr<float>(), x.data_ptr<float>(),
                               dx.data_ptr<float>(), rows, F, cur_stream());
  return dx;
}

Tensor rope_fwd(const Tensor& x, const Tensor& cos_t, const Tensor& sin_t,
                bool conj) {
  CHECK_GPU(x);
  TORCH_CHECK(x.dim() == 4, "rope: x must be [s,b,h,d]");
  const int d = x.size(3);
  TORCH_CHECK(d % 16 == 0, "rope: head dim must be %16");
  const long rows = x.numel() / d;
  const int bh = x.size(1) * x.size(2);
  auto c = cos_t.contiguous().to(at::kFloat);
  auto s = sin_t.contiguous().to(at::kFloat);
  auto y = at::empty_like(x);
  if (is_bf16(x))
    rope_launch_t<__bf16>(bfp(x), bfp_mut(y), c.data_ptr<float>(),
                          s.data_ptr<float>(), rows, bh, d, conj,
                          cur_stream());
  else
    rope_launch_t<float>(x.data_ptr<float>(), y.data_ptr<float>(),
                         c.data_ptr<float>(), s.data_ptr<float>(),
                         rows, bh, d, conj, cur_stream());
  return y;
}

// ---- flash attention ------------------------------------------------------
std::tuple<Tensor, Tensor> flash_attn_fwd(
    const Tensor& q, const Tensor& k, const Tensor& v, bool causal,
    double scale, const c10::optional<Tensor>& bias = c10::nullopt,
    bool sbhd = false, int64_t window = 0) {
  TORCH_CHECK(window == 0 || causal,
              "flash_attn: sliding window requires causal");
  CHECK_GPU(q); CHECK_GPU(k); CHECK_GPU(v);
  TORCH_CHECK(is_bf16(q), "flash_attn: bf16 only on the native path");
  TORCH_CHECK(q.dim() == 4, "flash_attn: q must be [b,s,h,d] (or [s,b,h,d] with sbhd)");
  const int b = sbhd ? q.size(1) : q.size(0);
  const int sq = sbhd ? q.size(0) : q.size(1);
  const int hq = q.size(2), d = q.size(3);
  const int skv = sbhd ? k.size(0) : k.size(1);
  const int hkv = k.size(2);
  TORCH_CHECK(d == 64 || d == 128, "flash_attn: head dim must be 64|128");
  TORCH_CHECK(hq % hkv == 0, "flash_attn: GQA needs hq % hkv == 0");
  const __bf16* bp = nullptr;
  Tensor bias_c;
  if (bias.has_value()) {
    bias_c = bias->contiguous();
    TORCH_CHECK(is_bf16(bias_c) && bias_c.dim() == 3 &&
                bias_c.size(0) == hq && bias_c.size(1) == sq &&
                bias_c.size(2) == skv,
                "flash_attn: bias must be bf16 [hq, sq, skv]");
    bp = bfp(bias_c);
  }
  auto o = at::empty_like(q);
  auto lse = at::empty({b, hq, sq}, q.options().dtype(at::kFloat));
  flash_fwd_launch(bfp(q), bfp(k), bfp(v), bfp_mut(o),
                   lse.data_ptr<float>(), b, sq, skv, hq, hkv, d,
                   (float)scale, causal, cur_stream(), bp, sbhd,
                   (int)window);
  return {o, lse};
}

std::vector<Tensor> flash_attn_bwd(
    const Tensor& dout, const Tensor& q, const Tensor& k, const Tensor& v,
    const Tensor& o, const Tensor& lse, bool causal, double scale,
    const c10::optional<Tensor>& bias = c10::nullopt, bool sbhd = false,
    int64_t window = 0) {
  TORCH_CHECK(window == 0 || causal,
              "flash_attn_bwd: sliding window requires causal");
  CHECK_GPU(dout); CHECK_GPU(q); CHECK_GPU(k); CHECK_GPU(v); CHECK_GPU(o);
  const int b = sbhd ? q.size(1) : q.size(0);
  const int sq = sbhd ? q.size(0) : q.size(1);
  const int hq = q.size(2), d = q.size(3);
  const int skv = sbhd ? k.size(0) : k.size(1);
  const int hkv = k.size(2);
  auto lsec = lse.contiguous();
  auto di = at::empty({b, hq, sq}, q.options().dtype(at::kFloat));
  attn_di_launch(bfp(dout), bfp(o), di.data_ptr<float>(), b, sq, hq, d,
                 cur_stream(), sbhd);
  auto dq = at::empty_like(q);
  auto dk_exp = sbhd ? at::empty({skv, b, hq, d}, k.options())
                     : at::empty({b, skv, hq, d}, k.options());
  auto dv_exp = sbhd ? at::empty({skv, b, hq, d}, v.options())
                     : at::empty({b, skv, hq, d}, v.options());
  const __bf16* bp = nullptr;
  float* dbp = nullptr;
  Tensor bias_c, dbias;
  if (bias.has_value()) {
    bias_c = bias->contiguous();
    TORCH_CHECK(is_bf16(bias_c) && bias_c.dim() == 3 &&
                bias_c.size(0) == hq && bias_c.size(1) == sq &&
                bias_c.size(2) == skv,
                "flash_attn_bwd: bias must be bf16 [hq, sq, skv]");
    bp = bfp(bias_c);
    dbias = at::zeros({hq, sq, skv}, q.options().dtype(at::kFloat));
    dbp = dbias.data_ptr<float>();
  }
  flash_bwd_launch(bfp(dout), bfp(q), bfp(k), bfp(v),
                   lsec.data_ptr<float>(), di.data_ptr<float>(),
                   bfp_mut(dq), bfp_mut(dk_exp), bfp_mut(dv_exp), b, sq, skv,
                   hq, hkv, d, (float)scale, causal, cur_stream(), bp, dbp,
                   sbhd, (int)window);
  Tensor dk = dk_exp, dv = dv_exp;
  if (hq != hkv) {
    const int rep = hq / hkv;
    if (sbhd) {
      dk = dk_exp.view({skv, b, hkv, rep, d}).sum(3);
      dv = dv_exp.view({skv, b, hkv, rep, d}).sum(3);
    } else {
      dk = dk_exp.view({b, skv, hkv, rep, d}).sum(3);
      dv = dv_exp.view({b, skv, hkv, rep, d}).sum(3);
    }
  }
  if (bias.has_value())
    return {dq, dk, dv, dbias};
  return {dq, dk, dv};
}

// Single-token decode attention against a KV cache (serving path).
// q: [b,hq,d]; k_cache/v_cache: [b,max_s,hkv,d]; attends to [0, cur_len).
Tensor decode_attn(const Tensor& q, const Tensor& k_cache,
                   const Tensor& v_cache, int64_t cur_len, double scale,
                   int64_t start = 0) {
  CHECK_GPU(q); CHECK_GPU(k_cache); CHECK_GPU(v_cache);
  TORCH_CHECK(is_bf16(q), "decode_attn: bf16 only");
  TORCH_CHECK(q.dim() == 3 && k_cache.dim() == 4, "decode_attn shapes");
  const int b = q.size(0), hq = q.size(1), d = q.size(2);
  const int max_s = k_cache.size(1), hkv = k_cache.size(2);
  TORCH_CHECK(d == 64 || d == 128, "decode_attn: head dim must be 64|128");
  TORCH_CHECK(cur_len >= 1 && cur_len <= max_s, "decode_attn: bad cur_len");
  TORCH_CHECK(start >= 0 && start < cur_len, "decode_attn: bad start");
  // windowed decode (mistral): positions [start, cur_len).  The kernels
  // are unchanged — advancing the cache pointers by start rows shifts
  // every (b, h) address uniformly within its batch row.
  const __bf16* kp = bfp(k_cache) + (long)start * hkv * d;
  const __bf16* vp = bfp(v_cache) + (long)start * hkv * d;
  cur_len -= start;
  auto o = at::empty_like(q);
  // split-KV: fill >=512 workgroups (2/CU) when b*hq alone can't, but keep
  // chunks >=128 positions so the merge stays cheap
  int n_chunks = b * hq >= 256 ? 1 : std::max(1, 512 / (b * hq));
  n_chunks = std::min<int>(n_chunks, (int)((cur_len + 127) / 128));
  // LDS cap: the weight buffer is cur_len/chunk fp32 — keep chunks <= 8192
  n_chunks = std::max<int>(n_chunks, (int)((cur_len + 8191) / 8192));
  Tensor ws;
  float* ws_ptr = nullptr;
  if (n_chunks > 1) {
    ws = at::empty({(long)b * hq * n_chunks * (d + 2)},
                   q.options().dtype(at::kFloat));
    ws_ptr = ws.data_ptr<float>();
  }
  decode_attn_launch(bfp(q), kp, vp, bfp_mut(o), ws_ptr,
                     n_chunks, b, hq, hkv, max_s, (int)cur_len, d,
                     (float)scale, nullptr, cur_stream());
  return o;
}

// hipGraph-capturable variant: the attended length lives in device memory
// (cur_len_dev int32 [1]); launch geometry + LDS are sized for max_len so
// a captured graph replays correctly as the cache grows.
Tensor decode_attn_graph(const Tensor& q, const Tensor& k_cache,
                         const Tensor& v_cache, const Tensor& cur_len_dev,
                         int64_t max_len, double scale) {
  CHECK_GPU(q); CHECK_GPU(cur_len_dev);
  TORCH_CHECK(is_bf16(q) && q.dim() == 3, "decode_attn_graph: bf16 [b,hq,d]");
  TORCH_CHECK(cur_len_dev.scalar_type() == at::kInt, "cur_len_dev: int32");
  const int b = q.size(0), hq = q.size(1), d = q.size(2);
  const int max_s = k_cache.size(1), hkv = k_cache.size(2);
  TORCH_CHECK(d == 64 || d == 128, "decode_attn_graph: head dim 64|128");
  TORCH_CHECK(max_len >= 1 && max_len <= max_s, "bad max_len");
  auto o = at::empty_like(q);
  int n_chunks = b * hq >= 256 ? 1 : std::max(1, 512 / (b * hq));
  n_chunks = std::min<int>(n_chunks, (int)((max_len + 127) / 128));
  n_chunks = std::max<int>(n_chunks, (int)((max_len + 8191) / 8192));
  auto ws = at::empty({(long)b * hq * n_chunks * (d + 2)},
                      q.options().dtype(at::kFloat));
  decode_attn_launch(bfp(q), bfp(k_cache), bfp(v_cache), bfp_mut(o),
                     ws.data_ptr<float>(), n_chunks, b, hq, hkv, max_s,
                     (int)max_len, d, (float)scale,
                     cur_len_dev.data_ptr<int>(), cur_stream());
  return o;
}

Tensor mfma_probe(const Tensor& A, const Tensor& B, bool alt) {
  CHECK_GPU(A); CHECK_GPU(B);
  auto D = at::zeros({32, 32}, A.options().dtype(at::kFloat));
  mfma_probe_launch(bfp(A), bfp(B), D.data_ptr<float>(), alt, cur_stream());
  return D;
}

// ---- vocab-parallel CE ----------------------------------------------------
Tensor ce_max(const Tensor& logits) {
  CHECK_GPU(logits);
  const long V = logits.size(-1);
  const long n = logits.numel() / V;
  auto out = at::empty({n}, logits.options().dtype(at::kFloat));
  if (is_bf16(logits))
    ce_max_launch_t<__bf16>(bfp(logits), out.data_ptr<float>(), n, V,
                            cur_stream());
  else
    ce_max_launch_t<float>(logits.data_ptr<float>(),
                           out.data_ptr<float>(), n, V, cur_stream());
  return out;
}

std::tuple<Tensor, Tensor> ce_sum_target(const Tensor& logits,
                                         const Tensor& target,
                                         const Tensor& gmax,
                                         long vocab_start) {
  CHECK_GPU(logits);
  const long V = logits.size(-1);
  const long n = logits.numel() / V;
  auto t = target.contiguous();
  auto g = gmax.contiguous();
  auto sumexp = at::empty({n}, logits.options().dtype(at::kFloat));
  auto tlogit = at::empty({n}, logits.options().dtype(at::kFloat));
  if (is_bf16(logits))
    ce_sum_target_launch_t<__bf16>(bfp(logits), t.data_ptr<long>(),
                                   g.data_ptr<float>(),
                                   sumexp.data_ptr<float>(),
                                   tlogit.data_ptr<float>(), n, V,
                                   vocab_start, cur_stream());
  else
    ce_sum_target_launch_t<float>(logits.data_ptr<float>(),
                                  t.data_ptr<long>(), g.data_ptr<float>(),
                                  sumexp.data_ptr<float>(),
                                  tlogit.data_ptr<float>(), n, V, vocab_start,
                                  cur_stream());
  return {sumexp, tlogit};
}

Tensor ce_bwd(Tensor logits, const Tensor& target, const Tensor& gmax,
              const Tensor& sumexp, const Tensor& grad_out, long vocab_start) {
  CHECK_GPU(logits);
  const long V = logits.size(-1);
  const long n = logits.numel() / V;
  auto t = target.contiguous();
  auto g = gmax.contiguous();
  auto se = sumexp.contiguous();
  auto go = grad_out.contiguous().to(at::kFloat);
  if (is_bf16(logits))
    ce_bwd_launch_t<__bf16>(bfp_mut(logits), t.data_ptr<long>(),
                            g.data_ptr<float>(), se.data_ptr<float>(),
                            go.data_ptr<float>(), n, V, vocab_start,
                            cur_stream());
  else
    ce_bwd_launch_t<float>(logits.data_ptr<float>(), t.data_ptr<long>(),
                           g.data_ptr<float>(), se.data_ptr<float>(),
                           go.data_ptr<float>(), n, V, vocab_start,
                           cur_stream());
  return logits;
}

// ---- grad accumulation ----------------------------------------------------
void grad_accum(Tensor flat, const Tensor& grad, long offset) {
  CHECK_GPU(flat); CHECK_GPU(grad);
  TORCH_CHECK(flat.scalar_type() == at::kFloat, "grad_accum: flat must be fp32");
  const long n = grad.numel();
  TORCH_CHECK(offset + n <= flat.numel(), "grad_accum: out of range");
  float* fp = flat.data_ptr<float>() + offset;
  if (is_bf16(grad))
    grad_accum_launch_t<__bf16>(fp, bfp(grad), n, cur_stream());
  else
    grad_accum_launch_t<float>(fp, grad.data_ptr<float>(), n, cur_stream());
}

// ---- MoE permute / unpermute ----------------------------------------------
Tensor moe_permute(const Tensor& x, const Tensor& rows) {
  CHECK_GPU(x); CHECK_GPU(rows);
  const long m = rows.numel();
  const int h = x.size(-1);
  TORCH_CHECK(h % 8 == 0, "moe_permute: h % 8");
  auto y = at::empty({m, (long)h}, x.options());
  if (m == 0) return y;
  if (is_bf16(x))
    moe_permute_launch_bf16_t(bfp(x), bfp_mut(y), rows.data_ptr<long>(), m,
                              h, cur_stream());
  else
    moe_permute_launch_f32_t(x.data_ptr<float>(), y.data_ptr<float>(),
                             rows.data_ptr<long>(), m, h, cur_stream());
  return y;
}

Tensor moe_permute_bwd(const Tensor& dy, const Tensor& rows, long n) {
  CHECK_GPU(dy);
  const long m = rows.numel();
  const int h = dy.size(-1);
  auto acc = at::zeros({n, (long)h}, dy.options().dtype(at::kFloat));
  if (m > 0) {
    if (is_bf16(dy))
      moe_permute_bwd_launch_bf16_t(bfp(dy), acc.data_ptr<float>(),
                                    rows.data_ptr<long>(), m, h,
                                    cur_stream());
    else
      moe_permute_bwd_launch_f32_t(dy.data_ptr<float>(),
                                   acc.data_ptr<float>(),
                                   rows.data_ptr<long>(), m, h,
                                   cur_stream());
  }
  return acc.to(dy.scalar_type());
}

Tensor moe_unpermute(const Tensor& back, const Tensor& probs,
                     const Tensor& inv_order, long n, long k) {
  CHECK_GPU(back);
  const int h = back.size(-1);
  auto out = at::empty({n, (long)h}, back.options());
  if (n == 0) return out;
  auto p = probs.contiguous().to(at::kFloat);
  if (is_bf16(back))
    moe_unpermute_launch_bf16_t(bfp(back), p.data_ptr<float>(),
                                inv_order.data_ptr<long>(), bfp_mut(out), n,
                                (int)k, h, cur_stream());
  else
    moe_unpermute_launch_f32_t(back.data_ptr<float>(), p.data_ptr<float>(),
                               inv_order.data_ptr<long>(),
                               out.data_ptr<float>(), n, (int)k, h,
                               cur_stream());
  return out;
}

std::tuple<Tensor, Tensor> moe_unpermute_bwd(const Tensor& dout,
                                             const Tensor& back,
                                             const Tensor& probs,
                                             const Tensor& t_of) {
  CHECK_GPU(dout); CHECK_GPU(back);
  const long m = back.size(0);
  const int h = back.size(-1);
  auto dback = at::empty_like(back);
  auto dprobs = at::zeros({m}, back.options().dtype(at::kFloat));
  auto p = probs.contiguous().to(at::kFloat);
  if (m > 0) {
    if (is_bf16(back))
      moe_unpermute_bwd_launch_bf16_t(bfp(dout), bfp(back),
                                      p.data_ptr<float>(),
                                      t_of.data_ptr<long>(), bfp_mut(dback),
                                      dprobs.data_ptr<float>(), m, h,
                                      cur_stream());
    else
      moe_unpermute_bwd_launch_f32_t(dout.data_ptr<float>(),
                                     back.data_ptr<float>(),
                                     p.data_ptr<float>(),
                                     t_of.data_ptr<long>(),
                                     dback.data_ptr<float>(),
                                     dprobs.data_ptr<float>(), m, h,
                                     cur_stream());
  }
  return {dback, dprobs};
}

// ---- grouped GEMM (MoE experts) -------------------------------------------
Tensor grouped_gemm(const Tensor& A, const Tensor& W, const Tensor& tiles,
                    long n_out, bool trans_b) {
  CHECK_GPU(A); CHECK_GPU(W); CHECK_GPU(tiles);
  TORCH_CHECK(is_bf16(A) && is_bf16(W), "grouped_gemm: bf16 only");
  const long M = A.size(0), K = A.size(1);
  TORCH_CHECK(K % 32 == 0 && n_out % 128 == 0,
              "grouped_gemm: K%32, N%128 required");
  auto C = at::empty({M, n_out}, A.options());
  if (M > 0)
    grouped_gemm_launch(bfp(A), bfp(W), bfp_mut(C), tiles.data_ptr(),
                        (int)tiles.size(0), (int)K, (int)n_out,
                        W.size(1) * W.size(2), trans_b, cur_stream());
  return C;
}

Tensor grouped_gemm_dw(const Tensor& A, const Tensor& dC,
                       const Tensor& row_off, long E) {
  CHECK_GPU(A); CHECK_GPU(dC); CHECK_GPU(row_off);
  const long K = A.size(1), N = dC.size(1);
  TORCH_CHECK(K % 128 == 0 && N % 128 == 0,
              "grouped_gemm_dw: K%128 and N%128 required");
  auto dW = at::empty({E, K, N}, A.options().dtype(at::kFloat));
  grouped_gemm_dw_launch(bfp(A), bfp(dC), dW.data_ptr<float>(),
                         row_off.data_ptr<int>(), (int)E, (int)K, (int)N,
                         cur_stream());
  return dW;
}

// ---- fused AdamW ----------------------------------------------------------
void fused_adamw(std::vector<Tensor> masters, std::vector<Tensor> grads,
                 std::vector<Tensor> ms, std::vector<Tensor> vs,
                 std::vector<Tensor> outs, long step, double lr, double beta1,
                 double beta2, double eps, double wd, double gscale = 1.0) {
  auto st = cur_stream();
  for (size_t i = 0; i < masters.size(); ++i) {
    auto& m = masters[i];
    const long n = m.numel();
    if (n == 0) continue;
    CHECK_GPU(m);
    TORCH_CHECK(m.scalar_type() == at::kFloat, "adamw: master must be fp32");
    float* mp = m.data_ptr<float>();
    float* ma = ms[i].data_ptr<float>();
    float* va = vs[i].data_ptr<float>();
    const bool g_bf = is_bf16(grads[i]);
    const bool o_bf = is_bf16(outs[i]);
    if (g_bf && o_bf)
      adamw_launch_t<__bf16, __bf16>(mp, bfp(grads[i]), ma, va,
                                     bfp_mut(outs[i]), n, (int)step, lr,
                                     beta1, beta2, eps, wd, (float)gscale, st);
    else if (g_bf && !o_bf)
      adamw_launch_t<__bf16, float>(mp, bfp(grads[i]), ma, va,
                                    outs[i].data_ptr<float>(), n, (int)step,
                                    lr, beta1, beta2, eps, wd, (float)gscale, st);
    else if (!g_bf && o_bf)
      adamw_launch_t<float, __bf16>(mp, grads[i].data_ptr<float>(), ma,
                                    va, bfp_mut(outs[i]), n, (int)step, lr,
                                    beta1, beta2, eps, wd, (float)gscale, st);
    else
      adamw_launch_t<float, float>(mp, grads[i].data_ptr<float>(), ma,
                                   va, outs[i].data_ptr<float>(), n,
                                   (int)step, lr, beta1, beta2, eps, wd, (float)gscale, st);
  }
}

}  // namespace

// ---- multi-tensor grad-norm sum of squares --------------------------------
Tensor multi_sumsq(std::vector<Tensor> grads) {
  TORCH_CHECK(!grads.empty(), "multi_sumsq: empty list");
  auto dev = grads[0].device();
  constexpr long CHUNK = 1 << 20;
  std::vector<long> meta;
  for (auto& g : grads) {
    CHECK_GPU(g);
    TORCH_CHECK(g.scalar_type() == at::kFloat && g.is_contiguous(),
                "multi_sumsq: fp32 contiguous only");
    const long n = g.numel();
    const long base = (long)(intptr_t)g.data_ptr<float>();
    for (long off = 0; off < n; off += CHUNK) {
      meta.push_back(base);
      meta.push_back(off);
      meta.push_back(std::min(CHUNK, n - off));
    }
  }
  auto meta_t = at::from_blob(meta.data(), {(long)meta.size()},
                              at::TensorOptions().dtype(at::kLong))
                    .to(dev, /*non_blocking=*/false);
  auto out = at::zeros({1}, grads[0].options().dtype(at::kFloat));
  multi_sumsq_launch(meta_t.data_ptr<long>(), (int)(meta.size() / 3),
                     out.data_ptr<float>(), cur_stream());
  return out;
}

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
  m.def("rmsnorm_fwd", &rmsnorm_fwd);
  m.def("rmsnorm_bwd", &rmsnorm_bwd);
  m.def("add_rmsnorm_fwd", &add_rmsnorm_fwd);
  m.def("add_rmsnorm_bwd", &add_rmsnorm_bwd);
  m.def("layernorm_fwd", &layernorm_fwd);
  m.def("layernorm_bwd", &layernorm_bwd);
  m.def("swiglu_fwd", &swiglu_fwd);
  m.def("swiglu_bwd", &swiglu_bwd);
  m.def("rope_fwd", &rope_fwd);
  m.def("flash_attn_fwd", &flash_attn_fwd, py::arg("q"), py::arg("k"), py::arg("v"), py::arg("causal"), py::arg("scale"), py::arg("bias") = py::none(), py::arg("sbhd") = false, py::arg("window") = 0);
  m.def("flash_attn_bwd", &flash_attn_bwd, py::arg("dout"), py::arg("q"), py::arg("k"), py::arg("v"), py::arg("o"), py::arg("lse"), py::arg("causal"), py::arg("scale"), py::arg("bias") = py::none(), py::arg("sbhd") = false, py::arg("window") = 0);
  m.def("decode_attn", &decode_attn, py::arg("q"), py::arg("k_cache"), py::arg("v_cache"), py::arg("cur_len"), py::arg("scale"), py::arg("start") = 0);
  m.def("decode_attn_graph", &decode_attn_graph);
  m.def("mfma_probe", &mfma_probe);
  m.def("ce_max", &ce_max);
  m.def("ce_sum_target", &ce_sum_target);
  m.def("ce_bwd", &ce_bwd);
  m.def("fused_adamw", &fused_adamw, py::arg("masters"), py::arg("grads"),
        py::arg("ms"), py::arg("vs"), py::arg("outs"), py::arg("step"),
        py::arg("lr"), py::arg("beta1"), py::arg("beta2"), py::arg("eps"),
        py::arg("wd"), py::arg("gscale") = 1.0);
  m.def("multi_sumsq", &multi_sumsq);
  m.def("grad_accum", &grad_accum);
  m.def("grouped_gemm", &grouped_gemm);
  m.def("moe_permute", &moe_permute);
  m.def("moe_permute_bwd", &moe_permute_bwd);
  m.def("moe_unpermute", &moe_unpermute);
  m.def("moe_unpermute_bwd", &moe_unpermute_bwd);
  m.def("grouped_gemm_dw", &grouped_gemm_dw);
  m.attr("arch") = "gfx950";
}
