// Python bindings for the gfx950 kernel library (native HIP — no hipify).
//
// Tensor checking + allocation happens here; kernels live in *.hip files
// exposing extern "C" launchers over raw pointers + hipStream_t.
#include <torch/extension.h>

#include <ATen/ATen.h>
#include <c10/hip/HIPStream.h>
#include <hip/hip_runtime.h>

#include <vector>
#include <algorithm>

#define HIP_OK(expr)                                                       \
  do {                                                                     \
    hipError_t _e = (expr);                                                \
    TORCH_CHECK(_e == hipSuccess, "HIP error: ", hipGetErrorString(_e));   \
  } while (0)

extern "C" {
hipError_t launch_layer_norm_fwd(const void*, const void*, const void*,
                                 void*, float*, float*, int64_t, int64_t,
                                 float, hipStream_t);
hipError_t launch_layer_norm_bwd_dx(const void*, const void*, const void*,
                                    const float*, const float*, void*,
                                    int64_t, int64_t, hipStream_t);
hipError_t launch_layer_norm_bwd_dwdb(const void*, const void*, const float*,
                                      const float*, float*, float*, float*,
                                      float*, int64_t, int64_t, int,
                                      hipStream_t);
hipError_t launch_add_layer_norm_fwd(const void*, const void*, const void*,
                                     const void*, void*, void*, float*,
                                     float*, int64_t, int64_t, float,
                                     hipStream_t);
hipError_t launch_add_layer_norm_bwd_dx(const void*, const void*,
                                        const void*, const void*,
                                        const float*, const float*, void*,
                                        int64_t, int64_t, hipStream_t);
hipError_t launch_bias_gelu_fwd(const void*, const void*, void*, int64_t,
                                int64_t, hipStream_t);
hipError_t launch_bias_gelu_bwd(const void*, const void*, const void*, void*,
                                float*, float*, int64_t, int64_t, int,
                                hipStream_t);
hipError_t launch_colsum_partial(const void*, float*, int64_t, int64_t, int,
                                 hipStream_t);
hipError_t launch_adamw(const void*, const void*, int, float, float, float,
                        float, float, float, int, hipStream_t);
hipError_t launch_fp8_quantize(const void*, void*, const float*, float*,
                               int64_t, int, hipStream_t);
hipError_t launch_fp8_quantize_dual(const void*, void*, void*, const float*,
                                    float*, int64_t, int64_t, int,
                                    hipStream_t);
hipError_t launch_skinny_gemm(const void*, const void*, float*,
                              const void*, void*, const float*, int64_t,
                              int64_t, int64_t, int64_t, int, int,
                              hipStream_t);
hipError_t launch_ce_fwd(const void*, const int64_t*, float*, float*, int64_t,
                         int64_t, hipStream_t);
hipError_t launch_ce_bwd(const float*, const void*, const int64_t*,
                         const float*, void*, int64_t, int64_t, hipStream_t);
hipError_t launch_attn_fwd(const void*, const void*, const void*, void*,
                           float*, int64_t, int64_t, int64_t, int64_t,
                           int64_t, float, int, const float*, const int*,
                           const int64_t*, hipStream_t);
hipError_t launch_attn_fwd_sbuf(const void*, const void*, const void*,
                                void*, float*, int64_t, int64_t, int64_t,
                                int64_t, int64_t, float, int,
                                const int64_t*, hipStream_t);
hipError_t launch_attn_fwd_v2(const void*, const void*, const void*, void*,
                              float*, int64_t, int64_t, int64_t, int64_t,
                              int64_t, float, int, const int64_t*,
                              hipStream_t);
hipError_t launch_mfma_probe(const void*, const void*, float*, hipStream_t);
hipError_t launch_mfma_probe32(const void*, const void*, float*,
                               hipStream_t);
hipError_t launch_attn_fwd_ablate(const void*, const void*, const void*,
                                  void*, float*, int64_t, int64_t, int64_t,
                                  int64_t, int64_t, float, int,
                                  const int64_t*, int, hipStream_t);
hipError_t launch_attn_bwd_dkv_ablate(const void*, const void*,
                                      const void*, const void*,
                                      const float*, const float*, void*,
                                      void*, int64_t, int64_t, int64_t,
                                      int64_t, int64_t, float, int,
                                      const int64_t*, int, hipStream_t);
hipError_t launch_attn_bwd(const void*, const void*, const void*,
                           const void*, const void*, const float*, float*,
                           void*, void*, void*, int64_t, int64_t, int64_t,
                           int64_t, int64_t, float, int, const float*,
                           const int64_t*, hipStream_t);
}

namespace {

hipStream_t cur_stream() {
  return c10::hip::getCurrentHIPStream().stream();
}

void check_bf16_contig(const at::Tensor& t, const char* name) {
  TORCH_CHECK(t.is_cuda(), name, " must be on GPU");
  TORCH_CHECK(t.scalar_type() == at::kBFloat16, name, " must be bf16");
  TORCH_CHECK(t.is_contiguous(), name, " must be contiguous");
}

// Deterministic column-partial stripe count: enough (stripes x col-chunks)
// blocks to fill 256 CUs, bounded by the row count.
int stripes_for(int64_t N, int64_t cols) {
  int64_t chunks = std::max<int64_t>(1, cols / 256);
  int64_t p = std::max<int64_t>(64, 2048 / chunks);
  return (int)std::min<int64_t>({p, N, 512});
}

// bias_gelu's dbias partial reads 8 cols/thread: fewer column chunks,
// so more stripes keep >=2k blocks in flight
int stripes_for_vec8(int64_t N, int64_t cols) {
  int64_t chunks = std::max<int64_t>(1, cols / 2048);
  int64_t p = std::max<int64_t>(64, 2048 / chunks);
  return (int)std::min<int64_t>({p, N, 512});
}

// ------------------------------- LayerNorm -------------------------------

std::vector<at::Tensor> layer_norm_fwd(const at::Tensor& x,
                                       const at::Tensor& w,
                                       const at::Tensor& b, double eps) {
  check_bf16_contig(x, "x");
  int64_t H = x.size(-1);
  int64_t N = x.numel() / H;
  TORCH_CHECK(H % 8 == 0, "H must be a multiple of 8");
  auto y = at::empty_like(x);
  auto f32 = x.options().dtype(at::kFloat);
  auto mean = at::empty({N}, f32);
  auto rstd = at::empty({N}, f32);
  HIP_OK(launch_layer_norm_fwd(x.const_data_ptr(), w.const_data_ptr(),
                               b.const_data_ptr(), y.mutable_data_ptr(),
                               (float*)mean.mutable_data_ptr(),
                               (float*)rstd.mutable_data_ptr(), N, H,
                               (float)eps, cur_stream()));
  return {y, mean, rstd};
}

std::vector<at::Tensor> layer_norm_bwd(const at::Tensor& dy,
                                       const at::Tensor& x,
                                       const at::Tensor& w,
                                       const at::Tensor& mean,
                                       const at::Tensor& rstd) {
  check_bf16_contig(dy, "dy");
  int64_t H = x.size(-1);
  int64_t N = x.numel() / H;
  auto dx = at::empty_like(x);
  auto f32 = x.options().dtype(at::kFloat);
  auto dw = at::empty({H}, f32);
  auto db = at::empty({H}, f32);
  const int kStripes = stripes_for(N, H);
  auto dw_part = at::empty({kStripes, H}, f32);
  auto db_part = at::empty({kStripes, H}, f32);
  HIP_OK(launch_layer_norm_bwd_dx(
      dy.const_data_ptr(), x.const_data_ptr(), w.const_data_ptr(),
      (const float*)mean.const_data_ptr(), (const float*)rstd.const_data_ptr(),
      dx.mutable_data_ptr(), N, H, cur_stream()));
  HIP_OK(launch_layer_norm_bwd_dwdb(
      dy.const_data_ptr(), x.const_data_ptr(), (const float*)mean.const_data_ptr(),
      (const float*)rstd.const_data_ptr(), (float*)dw_part.mutable_data_ptr(),
      (float*)db_part.mutable_data_ptr(), (float*)dw.mutable_data_ptr(),
      (float*)db.mutable_data_ptr(), N, H, kStripes, cur_stream()));
  at::sum_out(dw, dw_part, {0});
  at::sum_out(db, db_part, {0});
  return {dx, dw, db};
}

// --------------------------- Add + LayerNorm -----------------------------

std::vector<at::Tensor> add_layer_norm_fwd(const at::Tensor& a,
                                           const c10::optional<at::Tensor>& b,
                                           const at::Tensor& w,
                                           const at::Tensor& bias,
                                           double eps) {
  check_bf16_contig(a, "a");
  int64_t H = a.size(-1);
  int64_t N = a.numel() / H;
  auto h = at::empty_like(a);
  auto y = at::empty_like(a);
  auto f32 = a.options().dtype(at::kFloat);
  auto mean = at::empty({N}, f32);
  auto rstd = at::empty({N}, f32);
  const void* bp = b.has_value() ? b->const_data_ptr() : nullptr;
  HIP_OK(launch_add_layer_norm_fwd(
      a.const_data_ptr(), bp, w.const_data_ptr(), bias.const_data_ptr(),
      h.mutable_data_ptr(), y.mutable_data_ptr(),
      (float*)mean.mutable_data_ptr(), (float*)rstd.mutable_data_ptr(), N,
      H, (float)eps, cur_stream()));
  return {h, y, mean, rstd};
}

std::vector<at::Tensor> add_layer_norm_bwd(
    const at::Tensor& dy, const c10::optional<at::Tensor>& dh,
    const at::Tensor& h, const at::Tensor& w, const at::Tensor& mean,
    const at::Tensor& rstd) {
  check_bf16_contig(dy, "dy");
  int64_t H = h.size(-1);
  int64_t N = h.numel() / H;
  auto dx = at::empty_like(h);
  auto f32 = h.options().dtype(at::kFloat);
  auto dw = at::empty({H}, f32);
  auto db = at::empty({H}, f32);
  const int kStripes = stripes_for(N, H);
  auto dw_part = at::empty({kStripes, H}, f32);
  auto db_part = at::empty({kStripes, H}, f32);
  const void* dhp = dh.has_value() ? dh->const_data_ptr() : nullptr;
  HIP_OK(launch_add_layer_norm_bwd_dx(
      dy.const_data_ptr(), dhp, h.const_data_ptr(), w.const_data_ptr(),
      (const float*)mean.const_data_ptr(),
      (const float*)rstd.const_data_ptr(), dx.mutable_data_ptr(), N, H,
      cur_stream()));
  HIP_OK(launch_layer_norm_bwd_dwdb(
      dy.const_data_ptr(), h.const_data_ptr(),
      (const float*)mean.const_data_ptr(),
      (const float*)rstd.const_data_ptr(),
      (float*)dw_part.mutable_data_ptr(), (float*)db_part.mutable_data_ptr(),
      (float*)dw.mutable_data_ptr(), (float*)db.mutable_data_ptr(), N, H,
      kStripes, cur_stream()));
  at::sum_out(dw, dw_part, {0});
  at::sum_out(db, db_part, {0});
  return {dx, dw, db};
}

// ------------------------------- BiasGelu --------------------------------

at::Tensor bias_gelu_fwd(const at::Tensor& x, const at::Tensor& bias) {
  check_bf16_contig(x, "x");
  int64_t F = x.size(-1);
  int64_t N = x.numel() / F;
  TORCH_CHECK(F % 8 == 0, "F must be a multiple of 8");
  auto y = at::empty_like(x);
  HIP_OK(launch_bias_gelu_fwd(x.const_data_ptr(), bias.const_data_ptr(),
                              y.mutable_data_ptr(), N, F, cur_stream()));
  return y;
}

std::vector<at::Tensor> bias_gelu_bwd(const at::Tensor& dy,
                                      const at::Tensor& x,
                                      const at::Tensor& bias) {
  check_bf16_contig(dy, "dy");
  int64_t F = x.size(-1);
  int64_t N = x.numel() / F;
  auto dx = at::empty_like(x);
  auto f32 = x.options().dtype(at::kFloat);
  const int kStripes = stripes_for_vec8(N, F);
  auto db = at::empty({F}, f32);
  auto db_part = at::empty({kStripes, F}, f32);
  HIP_OK(launch_bias_gelu_bwd(dy.const_data_ptr(), x.const_data_ptr(),
                              bias.const_data_ptr(), dx.mutable_data_ptr(),
                              (float*)db_part.mutable_data_ptr(),
                              (float*)db.mutable_data_ptr(), N, F, kStripes,
                              cur_stream()));
  at::sum_out(db, db_part, {0});
  return {dx, db};
}

// column sum of a bf16 [N, F] tensor -> fp32 [F] (striped partials +
// at::sum): the bias-gradient reduction for plain +bias layers (the
// at::native reduce path cost ~8 ms/step in the r2 profile)
at::Tensor colsum_bf16(const at::Tensor& t) {
  check_bf16_contig(t, "t");
  int64_t F = t.size(-1);
  int64_t N = t.numel() / F;
  auto f32 = t.options().dtype(at::kFloat);
  const int kStripes = stripes_for_vec8(N, F);
  auto part = at::empty({kStripes, F}, f32);
  auto out = at::empty({F}, f32);
  HIP_OK(launch_colsum_partial(t.const_data_ptr(),
                               (float*)part.mutable_data_ptr(), N, F,
                               kStripes, cur_stream()));
  at::sum_out(out, part, {0});
  return out;
}

// ----------------------------- skinny GEMM -------------------------------

// y[M,N] bf16 = x[M,K] @ Wp^T (+ bias) for decode-shaped M<=64; wp is
// the packed [K/8, N, 8] layout (bf16 or e4m3 — pass wscale for e4m3).
// Two launches: split-K partials then a deterministic finalize
// (reduce + bias + cast) — no zero-fill, no atomics.
at::Tensor skinny_gemm(const at::Tensor& wp, const at::Tensor& x,
                       const c10::optional<at::Tensor>& wscale,
                       const c10::optional<at::Tensor>& bias,
                       int64_t N, int64_t K, int64_t splits) {
  TORCH_CHECK(wp.is_cuda() && x.is_cuda() && x.is_contiguous());
  TORCH_CHECK(x.dim() == 2 && x.size(1) == K, "x must be [M, K]");
  const int64_t M = x.size(0);
  TORCH_CHECK(M >= 1 && M <= 64, "skinny path is for M <= 64");
  const bool fp8 = wscale.has_value();
  int64_t MT = 4;
  while (MT < M) MT *= 2;
  at::Tensor xp = x;
  if (MT != M) {
    xp = at::zeros({MT, K}, x.options());
    xp.narrow(0, 0, M).copy_(x);
  }
  auto part = at::empty({splits, MT, N}, x.options().dtype(at::kFloat));
  auto y = at::empty({M, N}, x.options());
  HIP_OK(launch_skinny_gemm(
      wp.const_data_ptr(), xp.const_data_ptr(),
      (float*)part.mutable_data_ptr(),
      bias.has_value() ? bias->const_data_ptr() : nullptr,
      y.mutable_data_ptr(),
      fp8 ? (const float*)wscale->const_data_ptr() : nullptr, M, MT, N, K,
      (int)splits, fp8 ? 1 : 0, cur_stream()));
  return y;
}

// -------------------------------- AdamW ----------------------------------

void adamw_step_raw(const at::Tensor& descs, const at::Tensor& chunks,
                    int64_t step, double lr, double beta1, double beta2,
                    double eps, double weight_decay, double grad_scale) {
  TORCH_CHECK(descs.is_cuda() && chunks.is_cuda());
  HIP_OK(launch_adamw(descs.const_data_ptr(), chunks.const_data_ptr(),
                      (int)chunks.size(0), (float)lr, (float)beta1,
                      (float)beta2, (float)eps, (float)weight_decay,
                      (float)grad_scale, (int)step, cur_stream()));
}

// ----------------------------- CrossEntropy ------------------------------

std::vector<at::Tensor> cross_entropy_fwd(const at::Tensor& logits,
                                          const at::Tensor& targets) {
  check_bf16_contig(logits, "logits");
  TORCH_CHECK(targets.scalar_type() == at::kLong);
  int64_t V = logits.size(-1);
  int64_t N = logits.numel() / V;
  TORCH_CHECK(V % 8 == 0, "V must be a multiple of 8");
  auto f32 = logits.options().dtype(at::kFloat);
  auto loss = at::empty({N}, f32);
  auto lse = at::empty({N}, f32);
  HIP_OK(launch_ce_fwd(logits.const_data_ptr(),
                       (const int64_t*)targets.const_data_ptr(),
                       (float*)loss.mutable_data_ptr(),
                       (float*)lse.mutable_data_ptr(), N, V, cur_stream()));
  return {loss, lse};
}

at::Tensor cross_entropy_bwd(const at::Tensor& dloss, const at::Tensor& logits,
                             const at::Tensor& targets,
                             const at::Tensor& lse) {
  check_bf16_contig(logits, "logits");
  int64_t V = logits.size(-1);
  int64_t N = logits.numel() / V;
  auto dlogits = at::empty_like(logits);
  auto dl = dloss.to(at::kFloat).contiguous();
  HIP_OK(launch_ce_bwd((const float*)dl.const_data_ptr(), logits.const_data_ptr(),
                       (const int64_t*)targets.const_data_ptr(),
                       (const float*)lse.const_data_ptr(),
                       dlogits.mutable_data_ptr(), N, V, cur_stream()));
  return dlogits;
}

// ------------------------------ Attention --------------------------------

void check_bf16_strided4(const at::Tensor& t, const char* name) {
  TORCH_CHECK(t.is_cuda(), name, " must be on GPU");
  TORCH_CHECK(t.scalar_type() == at::kBFloat16, name, " must be bf16");
  TORCH_CHECK(t.dim() == 4, name, " must be 4-D [B,H,S,D]");
  TORCH_CHECK(t.stride(3) == 1, name, " head-dim must be contiguous");
}

// q/k/v/o are logical [B, H, S, D] views with arbitrary B/H/S strides and a
// contiguous D — so the packed qkv layout [B, S, heads, 3*D] feeds the
// kernel directly, with no permute/contiguous copies on the hot path.

// ALiBi slopes: optional per-head fp32 device tensor [H]; nullptr = off
static const float* alibi_ptr(const c10::optional<at::Tensor>& alibi,
                              int64_t H) {
  if (!alibi.has_value() || !alibi->defined()) return nullptr;
  TORCH_CHECK(alibi->scalar_type() == at::kFloat &&
                  alibi->is_contiguous() && alibi->numel() == H,
              "alibi slopes must be contiguous fp32 [H]");
  return (const float*)alibi->const_data_ptr();
}

// kv_lens: optional per-batch int32 [B] real KV lengths (continuous
// batching decode); elements past a slot's length are masked
static const int* kv_lens_ptr(const c10::optional<at::Tensor>& kv_lens,
                              int64_t B) {
  if (!kv_lens.has_value() || !kv_lens->defined()) return nullptr;
  TORCH_CHECK(kv_lens->scalar_type() == at::kInt &&
                  kv_lens->is_contiguous() && kv_lens->numel() == B,
              "kv_lens must be contiguous int32 [B]");
  return (const int*)kv_lens->const_data_ptr();
}

at::Tensor attn_fwd_out(const at::Tensor& q, const at::Tensor& k,
                        const at::Tensor& v, at::Tensor o, at::Tensor lse,
                        bool causal, double scale,
                        const c10::optional<at::Tensor>& alibi =
                            c10::nullopt,
                        const c10::optional<at::Tensor>& kv_lens =
                            c10::nullopt) {
  check_bf16_strided4(q, "q");
  check_bf16_strided4(k, "k");
  check_bf16_strided4(v, "v");
  check_bf16_strided4(o, "o");
  int64_t B = q.size(0), H = q.size(1), S = q.size(2), D = q.size(3);
  int64_t Skv = k.size(2);
  TORCH_CHECK(D % 16 == 0 && D <= 256, "head_dim must be <=256, mult of 16");
  TORCH_CHECK(lse.is_contiguous() && lse.numel() == B * H * S);
  int64_t strides[12] = {q.stride(0), q.stride(1), q.stride(2),
                         k.stride(0), k.stride(1), k.stride(2),
                         v.stride(0), v.stride(1), v.stride(2),
                         o.stride(0), o.stride(1), o.stride(2)};
  HIP_OK(launch_attn_fwd(q.const_data_ptr(), k.const_data_ptr(),
                         v.const_data_ptr(), o.mutable_data_ptr(),
                         (float*)lse.mutable_data_ptr(), B, H, S, Skv, D,
                         (float)scale, causal ? 1 : 0, alibi_ptr(alibi, H),
                         kv_lens_ptr(kv_lens, B), strides, cur_stream()));
  return o;
}

at::Tensor attn_fwd_ablate(const at::Tensor& q, const at::Tensor& k,
                           const at::Tensor& v, bool causal, double scale,
                           int64_t abl) {
  int64_t B = q.size(0), H = q.size(1), S = q.size(2), D = q.size(3);
  auto o = at::empty_like(q);
  auto lse = at::empty({B, H, S}, q.options().dtype(at::kFloat));
  int64_t strides[12] = {q.stride(0), q.stride(1), q.stride(2),
                         k.stride(0), k.stride(1), k.stride(2),
                         v.stride(0), v.stride(1), v.stride(2),
                         o.stride(0), o.stride(1), o.stride(2)};
  HIP_OK(launch_attn_fwd_ablate(q.const_data_ptr(), k.const_data_ptr(),
                                v.const_data_ptr(), o.mutable_data_ptr(),
                                (float*)lse.mutable_data_ptr(), B, H, S,
                                k.size(2), D, (float)scale, causal ? 1 : 0,
                                strides, (int)abl, cur_stream()));
  return o;
}

std::vector<at::Tensor> attn_fwd_sbuf(const at::Tensor& q,
                                      const at::Tensor& k,
                                      const at::Tensor& v, bool causal,
                                      double scale) {
  int64_t B = q.size(0), H = q.size(1), S = q.size(2);
  auto o = at::empty_like(q.contiguous());
  auto lse = at::empty({B, H, S}, q.options().dtype(at::kFloat));
  int64_t strides[12] = {q.stride(0), q.stride(1), q.stride(2),
                         k.stride(0), k.stride(1), k.stride(2),
                         v.stride(0), v.stride(1), v.stride(2),
                         o.stride(0), o.stride(1), o.stride(2)};
  HIP_OK(launch_attn_fwd_sbuf(q.const_data_ptr(), k.const_data_ptr(),
                              v.const_data_ptr(), o.mutable_data_ptr(),
                              (float*)lse.mutable_data_ptr(), B, H, S,
                              k.size(2), q.size(3), (float)scale,
                              causal ? 1 : 0, strides, cur_stream()));
  return {o, lse};
}

std::vector<at::Tensor> attn_fwd_v2(const at::Tensor& q,
                                    const at::Tensor& k,
                                    const at::Tensor& v, bool causal,
                                    double scale) {
  int64_t B = q.size(0), H = q.size(1), S = q.size(2), D = q.size(3);
  auto o = at::empty_like(q.contiguous());
  auto lse = at::empty({B, H, S}, q.options().dtype(at::kFloat));
  int64_t strides[12] = {q.stride(0), q.stride(1), q.stride(2),
                         k.stride(0), k.stride(1), k.stride(2),
                         v.stride(0), v.stride(1), v.stride(2),
                         o.stride(0), o.stride(1), o.stride(2)};
  HIP_OK(launch_attn_fwd_v2(q.const_data_ptr(), k.const_data_ptr(),
                            v.const_data_ptr(), o.mutable_data_ptr(),
                            (float*)lse.mutable_data_ptr(), B, H, S,
                            k.size(2), D, (float)scale, causal ? 1 : 0,
                            strides, cur_stream()));
  return {o, lse};
}

std::vector<at::Tensor> attn_fwd(const at::Tensor& q, const at::Tensor& k,
                                 const at::Tensor& v, bool causal,
                                 double scale,
                                 const c10::optional<at::Tensor>& alibi =
                                     c10::nullopt,
                                 const c10::optional<at::Tensor>& kv_lens =
                                     c10::nullopt) {
  int64_t B = q.size(0), H = q.size(1), S = q.size(2);
  auto o = at::empty_like(q.contiguous());
  auto lse = at::empty({B, H, S}, q.options().dtype(at::kFloat));
  attn_fwd_out(q, k, v, o, lse, causal, scale, alibi, kv_lens);
  return {o, lse};
}

// Blocked attention forward for head_dim > 128 (the fused kernel tiles
// head_dim in registers up to 128): q-block loop, scores via bf16
// hipBLASLt GEMMs, fp32 softmax, never materializes more than one
// [Bq, Skv] block.  Serves CodeGen-6B/16B-class models (head_dim 256).
std::vector<at::Tensor> attn_fwd_blocked(const at::Tensor& q_,
                                         const at::Tensor& k_,
                                         const at::Tensor& v_, bool causal,
                                         double scale) {
  auto q = q_.contiguous(), k = k_.contiguous(), v = v_.contiguous();
  int64_t B = q.size(0), H = q.size(1), S = q.size(2);
  int64_t Skv = k.size(2);
  auto o = at::empty_like(q);
  auto lse = at::empty({B, H, S}, q.options().dtype(at::kFloat));
  const int64_t BQ = 256;
  int64_t qoff = Skv - S;  // cached decode: q global position offset
  for (int64_t qs = 0; qs < S; qs += BQ) {
    int64_t qe = std::min(qs + BQ, S);
    auto qb = q.slice(2, qs, qe);
    int64_t ke = causal ? std::min(qe + qoff, Skv) : Skv;
    auto kb = k.slice(2, 0, ke);
    auto s_blk = at::matmul(qb, kb.transpose(-1, -2)).to(at::kFloat)
                     .mul_(scale);
    if (causal) {
      // mask j > i + qoff
      auto qi = at::arange(qs + qoff, qe + qoff, q.options().dtype(at::kLong));
      auto kj = at::arange(ke, q.options().dtype(at::kLong));
      auto mask = kj.view({1, 1, 1, ke}) > qi.view({1, 1, qe - qs, 1});
      s_blk.masked_fill_(mask, -std::numeric_limits<float>::infinity());
    }
    auto m = std::get<0>(s_blk.max(-1, true));
    auto p = at::exp(s_blk - m);
    auto l = p.sum(-1, true);
    o.slice(2, qs, qe).copy_(
        at::matmul((p / l).to(at::kBFloat16), v.slice(2, 0, ke)));
    lse.slice(2, qs, qe).copy_((m + at::log(l)).squeeze(-1));
  }
  return {o, lse};
}

// Attention backward: deterministic blocked recompute; all GEMMs in bf16
// (hipBLASLt / MFMA), softmax math in fp32.  The q-block loop keeps the
// S x S score matrix from materializing beyond one [Bq, Skv] block.
// (A fully hand-written HIP bwd kernel replaces this on the optimization
// path.)
std::vector<at::Tensor> attn_bwd_blocked(const at::Tensor& dout_, const at::Tensor& q_,
                                 const at::Tensor& k_, const at::Tensor& v_,
                                 const at::Tensor& o_, const at::Tensor& lse,
                                 bool causal, double scale) {
  auto q = q_.contiguous(), k = k_.contiguous(), v = v_.contiguous();
  auto dout = dout_.contiguous(), o = o_.contiguous();
  int64_t B = q.size(0), H = q.size(1), S = q.size(2), D = q.size(3);
  int64_t Skv = k.size(2);
  auto dq = at::empty_like(q);
  auto dk = at::zeros_like(k, k.options().dtype(at::kFloat));
  auto dv = at::zeros_like(v, v.options().dtype(at::kFloat));
  // delta = rowsum(dO * O), fp32, computed once
  auto delta = (dout.to(at::kFloat) * o.to(at::kFloat)).sum(-1, true);
  auto lse4 = lse.view({B, H, S, 1});
  const int64_t BQ = 256;
  for (int64_t qs = 0; qs < S; qs += BQ) {
    int64_t qe = std::min(qs + BQ, S);
    auto qb = q.slice(2, qs, qe);
    auto dob = dout.slice(2, qs, qe);
    int64_t ke = causal ? std::min(qe, Skv) : Skv;
    auto kb = k.slice(2, 0, ke);
    auto vb = v.slice(2, 0, ke);
    // p = exp(s*scale - lse), zeroed above the causal diagonal via tril
    auto s = at::matmul(qb, kb.transpose(-1, -2));  // bf16 GEMM
    auto p = at::exp(s.to(at::kFloat) * scale - lse4.slice(2, qs, qe));
    if (causal) p = p.tril_(qs);
    auto p_bf = p.to(at::kBFloat16);
    dv.slice(2, 0, ke).add_(at::matmul(p_bf.transpose(-1, -2), dob));
    auto dp = at::matmul(dob, vb.transpose(-1, -2));  // bf16 GEMM
    auto ds = p.mul_(dp.to(at::kFloat) - delta.slice(2, qs, qe))
                  .mul_(scale);  // reuse p storage
    auto ds_bf = ds.to(at::kBFloat16);
    dq.slice(2, qs, qe).copy_(at::matmul(ds_bf, kb));
    dk.slice(2, 0, ke).add_(at::matmul(ds_bf.transpose(-1, -2), qb));
  }
  return {dq, dk.to(k.scalar_type()), dv.to(v.scalar_type())};
}

// Hand-written flash-attention backward (gfx950 MFMA, deterministic
// two-kernel split).  All tensors are logical [B,H,S,D] views with
// arbitrary B/H/S strides (contiguous D) — dq/dk/dv may be strided views
// into a packed dqkv buffer.
void attn_bwd_out(const at::Tensor& dout, const at::Tensor& q,
                  const at::Tensor& k, const at::Tensor& v,
                  const at::Tensor& o, const at::Tensor& lse, at::Tensor dq,
                  at::Tensor dk, at::Tensor dv, bool causal, double scale,
                  const c10::optional<at::Tensor>& alibi = c10::nullopt) {
  check_bf16_strided4(q, "q");
  check_bf16_strided4(dout, "dout");
  check_bf16_strided4(dq, "dq");
  int64_t B = q.size(0), H = q.size(1), S = q.size(2), D = q.size(3);
  int64_t Skv = k.size(2);
  auto delta = at::empty({B, H, S}, q.options().dtype(at::kFloat));
  auto lse_c = lse.contiguous();
  int64_t strides[24] = {
      q.stride(0), q.stride(1), q.stride(2),
      k.stride(0), k.stride(1), k.stride(2),
      v.stride(0), v.stride(1), v.stride(2),
      dout.stride(0), dout.stride(1), dout.stride(2),
      dq.stride(0), dq.stride(1), dq.stride(2),
      dk.stride(0), dk.stride(1), dk.stride(2),
      dv.stride(0), dv.stride(1), dv.stride(2),
      o.stride(0), o.stride(1), o.stride(2)};
  HIP_OK(launch_attn_bwd(q.const_data_ptr(), k.const_data_ptr(),
                         v.const_data_ptr(), o.const_data_ptr(),
                         dout.const_data_ptr(),
                         (const float*)lse_c.const_data_ptr(),
                         (float*)delta.mutable_data_ptr(),
                         dq.mutable_data_ptr(), dk.mutable_data_ptr(),
                         dv.mutable_data_ptr(), B, H, S, Skv, D,
                         (float)scale, causal ? 1 : 0, alibi_ptr(alibi, H),
                         strides, cur_stream()));
}

std::vector<at::Tensor> attn_bwd(const at::Tensor& dout, const at::Tensor& q,
                                 const at::Tensor& k, const at::Tensor& v,
                                 const at::Tensor& o, const at::Tensor& lse,
                                 bool causal, double scale,
                                 const c10::optional<at::Tensor>& alibi =
                                     c10::nullopt) {
  auto dq = at::empty_like(q.contiguous());
  auto dk = at::empty_like(k.contiguous());
  auto dv = at::empty_like(v.contiguous());
  attn_bwd_out(dout, q, k, v, o, lse, dq, dk, dv, causal, scale, alibi);
  return {dq, dk, dv};
}

void attn_bwd_dkv_ablate(const at::Tensor& dout, const at::Tensor& q,
                         const at::Tensor& k, const at::Tensor& v,
                         const at::Tensor& lse, const at::Tensor& delta,
                         at::Tensor dk, at::Tensor dv, bool causal,
                         double scale, int64_t abl) {
  int64_t B = q.size(0), H = q.size(1), S = q.size(2), D = q.size(3);
  int64_t strides[24] = {
      q.stride(0), q.stride(1), q.stride(2),
      k.stride(0), k.stride(1), k.stride(2),
      v.stride(0), v.stride(1), v.stride(2),
      dout.stride(0), dout.stride(1), dout.stride(2),
      0, 0, 0,
      dk.stride(0), dk.stride(1), dk.stride(2),
      dv.stride(0), dv.stride(1), dv.stride(2), 0, 0, 0};
  HIP_OK(launch_attn_bwd_dkv_ablate(
      q.const_data_ptr(), k.const_data_ptr(), v.const_data_ptr(),
      dout.const_data_ptr(), (const float*)lse.const_data_ptr(),
      (const float*)delta.const_data_ptr(), dk.mutable_data_ptr(),
      dv.mutable_data_ptr(), B, H, S, k.size(2), D, (float)scale,
      causal ? 1 : 0, strides, (int)abl, cur_stream()));
}

at::Tensor mfma_probe(const at::Tensor& a, const at::Tensor& b) {
  check_bf16_contig(a, "a");
  auto d = at::empty({16, 16}, a.options().dtype(at::kFloat));
  HIP_OK(launch_mfma_probe(a.const_data_ptr(), b.const_data_ptr(),
                           (float*)d.mutable_data_ptr(), cur_stream()));
  return d;
}

at::Tensor mfma_probe32(const at::Tensor& a, const at::Tensor& b) {
  check_bf16_contig(a, "a");
  auto d = at::empty({32, 32}, a.options().dtype(at::kFloat));
  HIP_OK(launch_mfma_probe32(a.const_data_ptr(), b.const_data_ptr(),
                             (float*)d.mutable_data_ptr(), cur_stream()));
  return d;
}

}  // namespace

// fp8 quantize: one-pass amax + delayed-scale cast (optional transposed
// twin for the dW GEMM).  x: bf16 contiguous; scale: fp32 scalar tensor
// on device (previous-step scale); returns (q, qt | empty, amax).
std::vector<at::Tensor> fp8_quantize(const at::Tensor& x,
                                     const at::Tensor& scale, bool dual,
                                     bool e5m2) {
  TORCH_CHECK(x.is_cuda() && x.is_contiguous() &&
              x.scalar_type() == at::kBFloat16, "x must be bf16 contiguous");
  TORCH_CHECK(scale.is_cuda() && scale.scalar_type() == at::kFloat &&
              scale.numel() == 1, "scale must be a device fp32 scalar");
  auto qtype = e5m2 ? at::kFloat8_e5m2 : at::kFloat8_e4m3fn;
  auto stream = c10::hip::getCurrentHIPStream().stream();
  auto amax = at::zeros({1}, x.options().dtype(at::kFloat));
  auto q = at::empty(x.sizes(), x.options().dtype(qtype));
  if (!dual) {
    TORCH_CHECK(x.numel() % 8 == 0, "numel must be divisible by 8");
    HIP_OK(launch_fp8_quantize(x.const_data_ptr(), q.data_ptr(),
                               (const float*)scale.const_data_ptr(),
                               (float*)amax.data_ptr(), x.numel(),
                               e5m2 ? 1 : 0, stream));
    return {q, at::Tensor(), amax};
  }
  TORCH_CHECK(x.dim() == 2, "dual quantize needs a 2-D tensor");
  int64_t M = x.size(0), N = x.size(1);
  auto qt = at::empty({N, M}, x.options().dtype(qtype));
  HIP_OK(launch_fp8_quantize_dual(x.const_data_ptr(), q.data_ptr(),
                                  qt.data_ptr(),
                                  (const float*)scale.const_data_ptr(),
                                  (float*)amax.data_ptr(), M, N,
                                  e5m2 ? 1 : 0, stream));
  return {q, qt, amax};
}

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
  // Bump when the Python<->extension call surface changes; checked by
  // alpa_amd/version.py (reference check_alpa_jaxlib_version, version.py:10)
  m.attr("ABI_VERSION") = 2;  // r2: fp8_quantize, NT-templated attention
  m.def("layer_norm_fwd", &layer_norm_fwd, "LayerNorm forward (gfx950)");
  m.def("layer_norm_bwd", &layer_norm_bwd, "LayerNorm backward (gfx950)");
  m.def("add_layer_norm_fwd", &add_layer_norm_fwd,
        "fused residual-add + LayerNorm fwd (gfx950)");
  m.def("add_layer_norm_bwd", &add_layer_norm_bwd,
        "fused residual-add + LayerNorm bwd (gfx950)");
  m.def("bias_gelu_fwd", &bias_gelu_fwd, "bias+GeLU forward (gfx950)");
  m.def("bias_gelu_bwd", &bias_gelu_bwd, "bias+GeLU backward (gfx950)");
  m.def("colsum_bf16", &colsum_bf16, "striped column sum (gfx950)");
  m.def("skinny_gemm", &skinny_gemm,
        "decode GEMV over packed weights (gfx950)");
  m.def("adamw_step_raw", &adamw_step_raw, "multi-tensor AdamW (gfx950)");
  m.def("fp8_quantize", &fp8_quantize,
        "one-pass fp8 amax+cast, optional transposed twin (gfx950)");
  m.def("cross_entropy_fwd", &cross_entropy_fwd, "fused CE fwd (gfx950)");
  m.def("cross_entropy_bwd", &cross_entropy_bwd, "fused CE bwd (gfx950)");
  m.def("attn_fwd", &attn_fwd, "flash attention fwd (gfx950 MFMA)",
        py::arg("q"), py::arg("k"), py::arg("v"), py::arg("causal"),
        py::arg("scale"), py::arg("alibi") = py::none(),
        py::arg("kv_lens") = py::none());
  m.def("attn_fwd_out", &attn_fwd_out, "flash attention fwd, strided out",
        py::arg("q"), py::arg("k"), py::arg("v"), py::arg("o"),
        py::arg("lse"), py::arg("causal"), py::arg("scale"),
        py::arg("alibi") = py::none(), py::arg("kv_lens") = py::none());
  m.def("attn_fwd_v2", &attn_fwd_v2, "32x32-MFMA fwd experiment");
  m.def("attn_fwd_ablate", &attn_fwd_ablate, "ablation variants (perf)");
  m.def("attn_bwd_dkv_ablate", &attn_bwd_dkv_ablate, "dkv ablation (perf)");
  m.def("attn_fwd_sbuf", &attn_fwd_sbuf,
        "single-buffer LDS fwd variant (2 blocks/CU experiment)");
  m.def("attn_fwd_blocked", &attn_fwd_blocked,
        "blocked fwd for head_dim > 128 (hipBLASLt scores)");
  m.def("attn_bwd", &attn_bwd, "flash attention bwd (gfx950 MFMA)",
        py::arg("dout"), py::arg("q"), py::arg("k"), py::arg("v"),
        py::arg("o"), py::arg("lse"), py::arg("causal"), py::arg("scale"),
        py::arg("alibi") = py::none());
  m.def("attn_bwd_out", &attn_bwd_out, "flash attention bwd, strided out",
        py::arg("dout"), py::arg("q"), py::arg("k"), py::arg("v"),
        py::arg("o"), py::arg("lse"), py::arg("dq"), py::arg("dk"),
        py::arg("dv"), py::arg("causal"), py::arg("scale"),
        py::arg("alibi") = py::none());
  m.def("attn_bwd_blocked", &attn_bwd_blocked,
        "attention bwd (blocked hipBLASLt reference)");
  m.def("mfma_probe", &mfma_probe, "MFMA 16x16x32 layout probe");
  m.def("mfma_probe32", &mfma_probe32, "MFMA 32x32x16 layout probe");
}
