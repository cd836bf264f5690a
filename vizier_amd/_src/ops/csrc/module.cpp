// Python bindings for the vizier_amd gfx950 kernels (torch extension).

#include <ATen/cuda/CUDAContext.h>
#include <torch/extension.h>

#include <algorithm>
#include <cstdlib>
#include <string>

#include <hip/hip_runtime.h>

extern "C" void launch_gram_matern52(const float* x1, const float* x2,
                                     const float* inv_ls, float* out,
                                     int n, int m, int d, float amp2,
                                     int sym, hipStream_t stream);

extern "C" void launch_gram_matern52_batched(
    const float* x, const float* inv_ls, const float* amp2,
    const float* noise, float* kout, float* gout, int r, int n, int d,
    hipStream_t stream);

extern "C" void launch_gram_matern52_bf16(
    const unsigned short* z1, const unsigned short* z2, const float* n1,
    const float* n2, float* out, int n, int m, int dp, float amp2,
    hipStream_t stream);

extern "C" void launch_gram_matern52_bf16_tiled(
    const unsigned short* z1, const unsigned short* z2, const float* n1,
    const float* n2, float* out, int n, int m, int dp, float amp2,
    hipStream_t stream);

extern "C" void launch_posterior_score(
    const float* xq, const float* x, const float* inv_ls,
    const float* alpha, const float* kinv, const unsigned char* onehot,
    float* out, int b, int n, int d, float amp2, float mean_c, int acq,
    float coef, float best_value, float tr_radius, hipStream_t stream);

extern "C" void launch_gram_matern52_fp8(
    const unsigned char* z1, const unsigned char* z2, const float* n1,
    const float* n2, float* out, int n, int m, int dp, float amp2,
    float scale2, hipStream_t stream);

extern "C" void launch_gram_matern52_fp8_tiled(
    const unsigned char* z1, const unsigned char* z2, const float* n1,
    const float* n2, float* out, int n, int m, int dp, float amp2,
    float scale2, hipStream_t stream);

extern "C" void launch_posterior_score_chunked(
    const float* xq, const float* x, const float* inv_ls,
    const float* alpha, const float* kinv, const unsigned char* onehot,
    float* k_ws, float* mu_ws, float* dist_ws, float* var_ws, float* out,
    int b, int n, int d, float amp2, float mean_c, int acq, float coef,
    float best_value, float tr_radius, hipStream_t stream);

extern "C" void launch_ps_kvec(
    const float* xq, const float* x, const float* inv_ls,
    const float* alpha, const unsigned char* onehot, float* k_ws,
    float* mu_ws, float* dist_ws, int b, int n, int d, float amp2,
    hipStream_t stream);

extern "C" void launch_ps_kvec_split(
    const float* xq, const float* x, const float* inv_ls,
    const float* alpha, const unsigned char* onehot, float* k_ws,
    float* mu_part, float* dist_part, float* mu_ws, float* dist_ws,
    int b, int n, int d, float amp2, int rchunks, hipStream_t stream);

extern "C" void launch_ps_quadform_tile(
    const float* k_ws, const float* kinv, float* var_part, float* quad,
    int b, int n, hipStream_t stream);

extern "C" void launch_ps_quadform_big(
    const float* k_ws, const float* kinv, float* part, float* quad,
    int b, int n, hipStream_t stream);

extern "C" void launch_ps_kvec_fp8(
    const float* z1f, const float* n1, const float* xq, const float* x,
    const unsigned char* z2q, const float* n2, const float* alpha,
    const unsigned char* onehot, float* k_ws, float* mu_ws,
    float* dist_ws, int b, int n, int d, int dp, float amp2,
    float scale, hipStream_t stream);

extern "C" void launch_ps_finalize_meanstd(
    const float* mu_ws, const float* var_ws, float* mean_out,
    float* sd_out, int b, float amp2, float mean_c, int nchunk,
    hipStream_t stream);

extern "C" void launch_ps_finalize_meanstd_direct(
    const float* mu_ws, const float* quad, float* mean_out,
    float* sd_out, int b, float amp2, float mean_c,
    hipStream_t stream);

extern "C" void launch_hv_scalarize_tr(
    const float* means, const float* sds, const float* weights,
    const float* ref, const float* dist, float* out, int b, int m,
    int s, float coef, float tr_radius, hipStream_t stream);

extern "C" void launch_ps_quadform_kernel_only(
    const float* k_ws, const float* kinv, float* var_ws, int b, int n,
    hipStream_t stream);

extern "C" void launch_ps_finalize_direct(
    const float* mu_ws, const float* dist_ws, const float* quad,
    float* out, int b, float amp2, float mean_c, int acq, float coef,
    float best_value, float tr_radius, hipStream_t stream);

extern "C" void launch_ps_kvec_bf16(
    const float* xq, const float* x, const unsigned short* z2b,
    const float* n2, const float* inv_ls, const float* alpha,
    const unsigned char* onehot, float* k_ws, float* mu_ws,
    float* dist_ws, int b, int n, int d, int dp, float amp2,
    hipStream_t stream);

extern "C" void launch_ps_quadform_finalize(
    const float* k_ws, const float* kinv, const float* mu_ws,
    const float* dist_ws, float* var_ws, float* out, int b, int n,
    float amp2, float mean_c, int acq, float coef, float best_value,
    float tr_radius, hipStream_t stream);

extern "C" void launch_batched_potrf(float* A, int* info, int r, int n,
                                     hipStream_t stream);
extern "C" int launch_batched_potrf_coop(float* A, int* info,
                                         unsigned int* bar, int r, int n,
                                         hipStream_t stream);
extern "C" int launch_batched_potrf_v5(const float* A, float* panels,
                                       float* dscratch, float* L,
                                       int* info, unsigned int* bar,
                                       int r, int n, hipStream_t stream);
extern "C" void launch_batched_trsv_lower(const float* L, float* b,
                                          int r, int n,
                                          hipStream_t stream);

extern "C" void launch_eagle_suggest(
    const float* pool_cont, const long* pool_cat, const float* rewards,
    const float* perturbations, const long* cat_sizes, float* out_cont,
    long* out_cat, const unsigned long long* iter_ptr, int n_batches,
    int batch_size, int pool_size, int q,
    int dc, int dcat, int max_cat, float visibility, float gravity,
    float neg_gravity, float norm_scale, float cat_factor, float p_same,
    unsigned long long seed, hipStream_t stream);

extern "C" int launch_eagle_sweep(
    float* pool_cont, float* rewards, float* perturbations,
    float* best_reward, unsigned long long* iter_ptr,
    unsigned int* barrier_buf, const float* x, const float* inv_ls,
    const float* alpha, const float* kinv, float* out_cont, float* k_ws,
    float* mu_ws, float* dist_ws, float* var_ws, int n_batches,
    int batch_size, int pool_size, int dc, int n, long long it_start,
    long long iterations, float visibility, float gravity,
    float neg_gravity, float norm_scale, float penalize_factor,
    float perturbation_lower_bound, float base_perturbation,
    unsigned long long seed_suggest, unsigned long long seed_update,
    float amp2, float mean_c, int acq, float coef, float best_value,
    float tr_radius, hipStream_t stream);

extern "C" void launch_eagle_update(
    float* pool_cont, long* pool_cat, float* rewards, float* perturbations,
    const float* batch_cont, const long* batch_cat,
    const float* batch_rewards, const long* cat_sizes, float* best_reward,
    unsigned long long* iter_ptr, int n_batches, int batch_size, int q,
    int dc, int dcat,
    float penalize_factor, float perturbation_lower_bound,
    float base_perturbation, unsigned long long seed,
    hipStream_t stream);

namespace {

torch::Tensor check_f32(const torch::Tensor& t, const char* name) {
  TORCH_CHECK(t.is_cuda(), name, " must be a GPU tensor");
  TORCH_CHECK(t.scalar_type() == torch::kFloat32, name, " must be fp32");
  return t.contiguous();
}

hipStream_t current_stream() {
  return at::cuda::getCurrentCUDAStream().stream();
}

torch::Tensor gram_matern52(torch::Tensor x1, torch::Tensor x2,
                            torch::Tensor lengthscales, double amplitude) {
  x1 = check_f32(x1, "x1");
  x2 = check_f32(x2, "x2");
  lengthscales = check_f32(lengthscales, "lengthscales");
  const int n = x1.size(0), m = x2.size(0), d = x1.size(1);
  TORCH_CHECK(x2.size(1) == d && lengthscales.numel() == d,
              "dimension mismatch");
  auto inv_ls = 1.0f / lengthscales;
  auto out = torch::empty({n, m}, x1.options());
  const int sym = (x1.data_ptr() == x2.data_ptr() && n == m) ? 1 : 0;
  launch_gram_matern52(x1.data_ptr<float>(), x2.data_ptr<float>(),
                       inv_ls.data_ptr<float>(), out.data_ptr<float>(), n,
                       m, d, (float)(amplitude * amplitude), sym,
                       current_stream());
  return out;
}

std::vector<torch::Tensor> gram_matern52_batched(
    torch::Tensor x, torch::Tensor lengthscales, torch::Tensor amplitude,
    torch::Tensor noise, bool want_grad_factor) {
  // (N, D) x; (R, D) lengthscales; (R,) amplitude, noise. Returns
  // [K (R, N, N) with noise*I folded in] or [K, G] with the
  // Matern-5/2 gradient factor.
  x = check_f32(x, "x");
  lengthscales = check_f32(lengthscales, "lengthscales");
  amplitude = check_f32(amplitude, "amplitude");
  noise = check_f32(noise, "noise");
  const int n = x.size(0), d = x.size(1);
  const int r = lengthscales.size(0);
  TORCH_CHECK(lengthscales.dim() == 2 && lengthscales.size(1) == d,
              "lengthscales must be (R, D)");
  TORCH_CHECK(amplitude.numel() == r && noise.numel() == r,
              "amplitude/noise must be (R,)");
  auto inv_ls = (1.0f / lengthscales).contiguous();
  auto amp2 = (amplitude * amplitude).contiguous();
  auto K = torch::empty({r, n, n}, x.options());
  torch::Tensor G;
  float* gptr = nullptr;
  if (want_grad_factor) {
    G = torch::empty({r, n, n}, x.options());
    gptr = G.data_ptr<float>();
  }
  launch_gram_matern52_batched(
      x.data_ptr<float>(), inv_ls.data_ptr<float>(),
      amp2.data_ptr<float>(), noise.data_ptr<float>(),
      K.data_ptr<float>(), gptr, r, n, d, current_stream());
  if (want_grad_factor) return {K, G};
  return {K};
}

torch::Tensor gram_matern52_bf16_impl(torch::Tensor x1, torch::Tensor x2,
                                      torch::Tensor lengthscales,
                                      double amplitude, bool tiled) {
  x1 = check_f32(x1, "x1");
  x2 = check_f32(x2, "x2");
  lengthscales = check_f32(lengthscales, "lengthscales");
  const int n = x1.size(0), m = x2.size(0), d = x1.size(1);
  TORCH_CHECK(x2.size(1) == d && lengthscales.numel() == d,
              "dimension mismatch");
  auto z1 = x1 / lengthscales;
  auto z2 = (x1.data_ptr() == x2.data_ptr()) ? z1 : x2 / lengthscales;
  const int dp = (d + 31) / 32 * 32;
  auto z1b = torch::zeros({n, dp},
                          x1.options().dtype(torch::kBFloat16));
  auto z2b = torch::zeros({m, dp},
                          x1.options().dtype(torch::kBFloat16));
  z1b.index_put_({torch::indexing::Slice(),
                  torch::indexing::Slice(0, d)}, z1.to(torch::kBFloat16));
  z2b.index_put_({torch::indexing::Slice(),
                  torch::indexing::Slice(0, d)}, z2.to(torch::kBFloat16));
  // Norms of the bf16-ROUNDED vectors: bf16 x bf16 products accumulate
  // exactly in fp32, so d^2 = n1 + n2 - 2 z1b.z2b is the exact squared
  // distance of the rounded inputs (diagonal exactly 0).
  auto n1 = (z1b.to(torch::kFloat32) * z1b.to(torch::kFloat32)).sum(-1);
  auto n2 = (z2b.to(torch::kFloat32) * z2b.to(torch::kFloat32)).sum(-1);
  auto out = torch::empty({n, m}, x1.options());
  auto launch = tiled ? launch_gram_matern52_bf16_tiled
                      : launch_gram_matern52_bf16;
  launch((const unsigned short*)z1b.data_ptr(),
         (const unsigned short*)z2b.data_ptr(), n1.data_ptr<float>(),
         n2.data_ptr<float>(), out.data_ptr<float>(), n, m, dp,
         (float)(amplitude * amplitude), current_stream());
  return out;
}

torch::Tensor gram_matern52_bf16(torch::Tensor x1, torch::Tensor x2,
                                 torch::Tensor lengthscales,
                                 double amplitude) {
  // LDS-tiled MFMA path once both dims fill 128x128 tiles; the
  // register-direct strip kernel has less overhead for small grams.
  const bool tiled = x1.size(0) >= 512 && x2.size(0) >= 512;
  return gram_matern52_bf16_impl(x1, x2, lengthscales, amplitude, tiled);
}

torch::Tensor gram_matern52_bf16_tiled(torch::Tensor x1, torch::Tensor x2,
                                       torch::Tensor lengthscales,
                                       double amplitude) {
  return gram_matern52_bf16_impl(x1, x2, lengthscales, amplitude, true);
}

torch::Tensor posterior_scores(torch::Tensor xq, torch::Tensor x,
                               torch::Tensor lengthscales, double amplitude,
                               double mean_c, torch::Tensor alpha,
                               torch::Tensor kinv, torch::Tensor onehot,
                               int64_t acq, double coef, double best_value,
                               double tr_radius) {
  xq = check_f32(xq, "xq");
  x = check_f32(x, "x");
  lengthscales = check_f32(lengthscales, "lengthscales");
  alpha = check_f32(alpha, "alpha");
  kinv = check_f32(kinv, "kinv");
  TORCH_CHECK(onehot.scalar_type() == torch::kUInt8 && onehot.is_cuda(),
              "onehot must be a uint8 GPU tensor");
  onehot = onehot.contiguous();
  const int b = xq.size(0), d = xq.size(1), n = x.size(0);
  TORCH_CHECK(d <= 512, "posterior_scores supports D <= 512");
  TORCH_CHECK(n <= 36000, "posterior_scores supports N <= 36000 (LDS)");
  TORCH_CHECK(x.size(1) == d && alpha.numel() == n &&
              kinv.size(0) == n && kinv.size(1) == n, "shape mismatch");
  auto inv_ls = 1.0f / lengthscales;
  auto out = torch::empty({b}, xq.options());
  launch_posterior_score(
      xq.data_ptr<float>(), x.data_ptr<float>(), inv_ls.data_ptr<float>(),
      alpha.data_ptr<float>(), kinv.data_ptr<float>(),
      onehot.data_ptr<unsigned char>(), out.data_ptr<float>(), b, n, d,
      (float)(amplitude * amplitude), (float)mean_c, (int)acq, (float)coef,
      (float)best_value, (float)tr_radius, current_stream());
  return out;
}

torch::Tensor gram_matern52_fp8_impl(torch::Tensor x1, torch::Tensor x2,
                                     torch::Tensor lengthscales,
                                     double amplitude, bool tiled) {
  x1 = check_f32(x1, "x1");
  x2 = check_f32(x2, "x2");
  lengthscales = check_f32(lengthscales, "lengthscales");
  const int n = x1.size(0), m = x2.size(0), d = x1.size(1);
  auto z1 = x1 / lengthscales;
  auto z2 = (x1.data_ptr() == x2.data_ptr()) ? z1 : x2 / lengthscales;
  // Pre-scale into comfortable e4m3 range (|z|/s <= ~8).
  double s = std::max(z1.abs().max().item<double>(),
                      z2.abs().max().item<double>()) / 8.0;
  s = std::max(s, 1e-8);
  const int dp = (d + 31) / 32 * 32;
  auto opts8 = x1.options().dtype(torch::kFloat8_e4m3fn);
  auto z1b = torch::zeros({n, dp}, opts8);
  auto z2b = torch::zeros({m, dp}, opts8);
  z1b.index_put_({torch::indexing::Slice(),
                  torch::indexing::Slice(0, d)},
                 (z1 / s).to(torch::kFloat8_e4m3fn));
  z2b.index_put_({torch::indexing::Slice(),
                  torch::indexing::Slice(0, d)},
                 (z2 / s).to(torch::kFloat8_e4m3fn));
  auto z1f = z1b.to(torch::kFloat32) * s;
  auto z2f = z2b.to(torch::kFloat32) * s;
  auto n1 = (z1f * z1f).sum(-1);
  auto n2 = (z2f * z2f).sum(-1);
  auto out = torch::empty({n, m}, x1.options());
  auto launch = tiled ? launch_gram_matern52_fp8_tiled
                      : launch_gram_matern52_fp8;
  launch((const unsigned char*)z1b.data_ptr(),
         (const unsigned char*)z2b.data_ptr(), n1.data_ptr<float>(),
         n2.data_ptr<float>(), out.data_ptr<float>(), n, m, dp,
         (float)(amplitude * amplitude), (float)(s * s),
         current_stream());
  return out;
}

torch::Tensor gram_matern52_fp8(torch::Tensor x1, torch::Tensor x2,
                                torch::Tensor lengthscales,
                                double amplitude) {
  const bool tiled = x1.size(0) >= 512 && x2.size(0) >= 512;
  return gram_matern52_fp8_impl(x1, x2, lengthscales, amplitude, tiled);
}

torch::Tensor gram_matern52_fp8_tiled(torch::Tensor x1, torch::Tensor x2,
                                      torch::Tensor lengthscales,
                                      double amplitude) {
  return gram_matern52_fp8_impl(x1, x2, lengthscales, amplitude, true);
}

namespace {

// Large-N variance quadform: ONE rocBLAS/hipBLASLt SGEMM by default.
// The hand-written split-K column-owner kernel (ps_quadform_big) is
// the VIZIER_AMD_QUADFORM=custom opt-in: it reads Kinv exactly once
// with perfect coalescing but measured 0.445 ms vs the library's
// 0.263 ms at (b=25, N=10^4) — the library's tiling wins this skinny
// shape (profiles/quadform_ab_r2.json); both are ~5x off the 50 us
// HBM floor, so the kernel stays for future split-K work.
// Small/medium-N quadform: the 64x64-tile kernel reads Kinv ONCE per
// call (vs once per candidate in the legacy chunked kernel — 32.6 us
// of a 58 us Eagle iteration at N=1000, profiles/sweep_kernels_r2.txt).
// b <= 32 only; returns the (b,) quadform.
torch::Tensor quadform_tile_small_n(const torch::Tensor& k_ws,
                                    const torch::Tensor& kinv, int b,
                                    int n, hipStream_t stream) {
  const int tiles_n = (n + 63) / 64;
  const long total = (long)tiles_n * tiles_n;
  auto var_part = torch::empty({b, total}, k_ws.options());
  auto quad = torch::empty({b}, k_ws.options());
  launch_ps_quadform_tile(k_ws.data_ptr<float>(),
                          kinv.data_ptr<float>(),
                          var_part.data_ptr<float>(),
                          quad.data_ptr<float>(), b, n, stream);
  return quad;
}

torch::Tensor quadform_large_n(const torch::Tensor& k_ws,
                               const torch::Tensor& kinv, int b, int n,
                               hipStream_t stream) {
  static const bool force_custom = []() {
    const char* s = getenv("VIZIER_AMD_QUADFORM");
    return s && std::string(s) == "custom";
  }();
  if (force_custom && b <= 32) {
    const int jchunks = (n + 255) / 256;
    int ichunks = (512 + jchunks - 1) / jchunks;
    if (ichunks < 1) ichunks = 1;
    auto part = torch::empty({b, (long)jchunks * ichunks},
                             k_ws.options());
    auto quad = torch::empty({b}, k_ws.options());
    launch_ps_quadform_big(k_ws.data_ptr<float>(),
                           kinv.data_ptr<float>(),
                           part.data_ptr<float>(),
                           quad.data_ptr<float>(), b, n, stream);
    return quad;
  }
  auto t = at::matmul(k_ws, kinv);
  return (k_ws * t).sum(-1);
}

}  // namespace

torch::Tensor gram_matern52_fp8_pre(
    torch::Tensor z1q, torch::Tensor z2q, torch::Tensor n1,
    torch::Tensor n2, double amplitude, double scale) {
  // Cross-gram from PRE-QUANTIZED fp8 operands: the per-call
  // host-side range scan (.item() sync) and conversions of the
  // training side are hoisted out by the caller (once per suggest),
  // so this is one launch and stream-capture-safe.
  TORCH_CHECK(z1q.scalar_type() == torch::kFloat8_e4m3fn &&
              z2q.scalar_type() == torch::kFloat8_e4m3fn,
              "z1q/z2q must be float8_e4m3fn");
  z1q = z1q.contiguous();
  z2q = z2q.contiguous();
  n1 = check_f32(n1, "n1");
  n2 = check_f32(n2, "n2");
  const int n = z1q.size(0), m = z2q.size(0), dp = z1q.size(1);
  TORCH_CHECK(z2q.size(1) == dp && dp % 32 == 0, "dp mismatch");
  auto out = torch::empty({n, m},
                          n1.options().dtype(torch::kFloat32));
  const bool tiled = n >= 512 && m >= 512;
  auto launch = tiled ? launch_gram_matern52_fp8_tiled
                      : launch_gram_matern52_fp8;
  launch((const unsigned char*)z1q.data_ptr(),
         (const unsigned char*)z2q.data_ptr(), n1.data_ptr<float>(),
         n2.data_ptr<float>(), out.data_ptr<float>(), n, m, dp,
         (float)(amplitude * amplitude), (float)(scale * scale),
         current_stream());
  return out;
}

torch::Tensor posterior_scores_chunked(
    torch::Tensor xq, torch::Tensor x, torch::Tensor lengthscales,
    double amplitude, double mean_c, torch::Tensor alpha,
    torch::Tensor kinv, torch::Tensor onehot, int64_t acq, double coef,
    double best_value, double tr_radius) {
  xq = check_f32(xq, "xq");
  x = check_f32(x, "x");
  lengthscales = check_f32(lengthscales, "lengthscales");
  alpha = check_f32(alpha, "alpha");
  kinv = check_f32(kinv, "kinv");
  onehot = onehot.contiguous();
  const int b = xq.size(0), d = xq.size(1), n = x.size(0);
  TORCH_CHECK(d <= 512, "posterior_scores supports D <= 512");
  auto inv_ls = 1.0f / lengthscales;
  auto k_ws = torch::empty({b, n}, xq.options());
  auto mu_ws = torch::empty({b}, xq.options());
  auto dist_ws = torch::empty({b}, xq.options());
  auto out = torch::empty({b}, xq.options());
  // Large-N quadform via one rocBLAS SGEMM: K_inv is read ONCE per
  // iteration instead of once per candidate (the per-candidate
  // streaming path moves b * N^2 * 4 bytes of HBM per call — 10 GB at
  // N=10^4, B=25 — and measured 4.2 ms/iter on BASELINE config 4).
  // Crossover default 4096 (below that K_inv is L2/MALL-resident and
  // the fused kernel wins on launch count); override with
  // VIZIER_AMD_PS_GEMM_N.
  static const int gemm_n_threshold = []() {
    const char* s = getenv("VIZIER_AMD_PS_GEMM_N");
    return s ? atoi(s) : 4096;
  }();
  if (n >= gemm_n_threshold) {
    const int rchunks = std::max(1, 512 / std::max(b, 1));
    auto mu_part = torch::empty({b, rchunks}, xq.options());
    auto dist_part = torch::empty({b, rchunks}, xq.options());
    launch_ps_kvec_split(
        xq.data_ptr<float>(), x.data_ptr<float>(),
        inv_ls.data_ptr<float>(), alpha.data_ptr<float>(),
        onehot.data_ptr<unsigned char>(), k_ws.data_ptr<float>(),
        mu_part.data_ptr<float>(), dist_part.data_ptr<float>(),
        mu_ws.data_ptr<float>(), dist_ws.data_ptr<float>(), b, n, d,
        (float)(amplitude * amplitude), rchunks, current_stream());
    auto quad = quadform_large_n(k_ws, kinv, b, n, current_stream());
    launch_ps_finalize_direct(
        mu_ws.data_ptr<float>(), dist_ws.data_ptr<float>(),
        quad.data_ptr<float>(), out.data_ptr<float>(), b,
        (float)(amplitude * amplitude), (float)mean_c, (int)acq,
        (float)coef, (float)best_value, (float)tr_radius,
        current_stream());
    return out;
  }
  if (b <= 32) {
    launch_ps_kvec(
        xq.data_ptr<float>(), x.data_ptr<float>(),
        inv_ls.data_ptr<float>(), alpha.data_ptr<float>(),
        onehot.data_ptr<unsigned char>(), k_ws.data_ptr<float>(),
        mu_ws.data_ptr<float>(), dist_ws.data_ptr<float>(), b, n, d,
        (float)(amplitude * amplitude), current_stream());
    auto quad = quadform_tile_small_n(k_ws, kinv, b, n,
                                      current_stream());
    launch_ps_finalize_direct(
        mu_ws.data_ptr<float>(), dist_ws.data_ptr<float>(),
        quad.data_ptr<float>(), out.data_ptr<float>(), b,
        (float)(amplitude * amplitude), (float)mean_c, (int)acq,
        (float)coef, (float)best_value, (float)tr_radius,
        current_stream());
    return out;
  }
  auto var_ws = torch::empty({b, 10}, xq.options());
  launch_posterior_score_chunked(
      xq.data_ptr<float>(), x.data_ptr<float>(), inv_ls.data_ptr<float>(),
      alpha.data_ptr<float>(), kinv.data_ptr<float>(),
      onehot.data_ptr<unsigned char>(), k_ws.data_ptr<float>(),
      mu_ws.data_ptr<float>(), dist_ws.data_ptr<float>(),
      var_ws.data_ptr<float>(), out.data_ptr<float>(), b, n, d,
      (float)(amplitude * amplitude), (float)mean_c, (int)acq,
      (float)coef, (float)best_value, (float)tr_radius,
      current_stream());
  return out;
}

torch::Tensor posterior_scores_bf16(
    torch::Tensor xq, torch::Tensor x, torch::Tensor z2b,
    torch::Tensor n2, torch::Tensor lengthscales, double amplitude,
    double mean_c, torch::Tensor alpha, torch::Tensor kinv,
    torch::Tensor onehot, int64_t acq, double coef, double best_value,
    double tr_radius) {
  // bf16 candidate grams against CACHED training operands (z2b =
  // bf16(x / lengthscales), n2 = rounded row norms — computed once per
  // suggest by ScoringFunction). Same 3-launch graph-capturable shape
  // as the fp32 chunked scorer.
  xq = check_f32(xq, "xq");
  x = check_f32(x, "x");
  n2 = check_f32(n2, "n2");
  lengthscales = check_f32(lengthscales, "lengthscales");
  alpha = check_f32(alpha, "alpha");
  kinv = check_f32(kinv, "kinv");
  TORCH_CHECK(z2b.scalar_type() == torch::kBFloat16 && z2b.is_cuda(),
              "z2b must be bf16 cuda");
  z2b = z2b.contiguous();
  onehot = onehot.contiguous();
  const int b = xq.size(0), d = xq.size(1), n = x.size(0);
  const int dp = z2b.size(1);
  TORCH_CHECK(dp % 32 == 0 && dp <= 512, "dp must be mult of 32, <=512");
  auto inv_ls = 1.0f / lengthscales;
  auto k_ws = torch::empty({b, n}, xq.options());
  auto mu_ws = torch::empty({b}, xq.options());
  auto dist_ws = torch::empty({b}, xq.options());
  auto out = torch::empty({b}, xq.options());
  launch_ps_kvec_bf16(
      xq.data_ptr<float>(), x.data_ptr<float>(),
      (const unsigned short*)z2b.data_ptr(), n2.data_ptr<float>(),
      inv_ls.data_ptr<float>(), alpha.data_ptr<float>(),
      onehot.data_ptr<unsigned char>(), k_ws.data_ptr<float>(),
      mu_ws.data_ptr<float>(), dist_ws.data_ptr<float>(), b, n, d, dp,
      (float)(amplitude * amplitude), current_stream());
  static const int gemm_n_threshold = []() {
    const char* s = getenv("VIZIER_AMD_PS_GEMM_N");
    return s ? atoi(s) : 4096;
  }();
  if (n >= gemm_n_threshold) {
    auto quad = quadform_large_n(k_ws, kinv, b, n, current_stream());
    launch_ps_finalize_direct(
        mu_ws.data_ptr<float>(), dist_ws.data_ptr<float>(),
        quad.data_ptr<float>(), out.data_ptr<float>(), b,
        (float)(amplitude * amplitude), (float)mean_c, (int)acq,
        (float)coef, (float)best_value, (float)tr_radius,
        current_stream());
    return out;
  }
  if (b <= 32) {
    auto quad = quadform_tile_small_n(k_ws, kinv, b, n,
                                      current_stream());
    launch_ps_finalize_direct(
        mu_ws.data_ptr<float>(), dist_ws.data_ptr<float>(),
        quad.data_ptr<float>(), out.data_ptr<float>(), b,
        (float)(amplitude * amplitude), (float)mean_c, (int)acq,
        (float)coef, (float)best_value, (float)tr_radius,
        current_stream());
    return out;
  }
  auto var_ws = torch::empty({b, 10}, xq.options());
  launch_ps_quadform_finalize(
      k_ws.data_ptr<float>(), kinv.data_ptr<float>(),
      mu_ws.data_ptr<float>(), dist_ws.data_ptr<float>(),
      var_ws.data_ptr<float>(), out.data_ptr<float>(), b, n,
      (float)(amplitude * amplitude), (float)mean_c, (int)acq,
      (float)coef, (float)best_value, (float)tr_radius,
      current_stream());
  return out;
}

std::vector<torch::Tensor> posterior_mean_std(
    torch::Tensor xq, torch::Tensor x, torch::Tensor lengthscales,
    double amplitude, double mean_c, torch::Tensor alpha,
    torch::Tensor kinv, torch::Tensor onehot) {
  // (mean, sd, min-Linf dist) for one GP over a candidate batch in
  // 3-4 launches — the per-metric piece of the fused MO scorer.
  xq = check_f32(xq, "xq");
  x = check_f32(x, "x");
  lengthscales = check_f32(lengthscales, "lengthscales");
  alpha = check_f32(alpha, "alpha");
  kinv = check_f32(kinv, "kinv");
  onehot = onehot.contiguous();
  const int b = xq.size(0), d = xq.size(1), n = x.size(0);
  TORCH_CHECK(d <= 512, "posterior_mean_std supports D <= 512");
  auto inv_ls = 1.0f / lengthscales;
  auto k_ws = torch::empty({b, n}, xq.options());
  auto mu_ws = torch::empty({b}, xq.options());
  auto dist_ws = torch::empty({b}, xq.options());
  auto mean = torch::empty({b}, xq.options());
  auto sd = torch::empty({b}, xq.options());
  const float amp2 = (float)(amplitude * amplitude);
  static const int gemm_n_threshold = []() {
    const char* s = getenv("VIZIER_AMD_PS_GEMM_N");
    return s ? atoi(s) : 4096;
  }();
  if (n >= gemm_n_threshold) {
    const int rchunks = std::max(1, 512 / std::max(b, 1));
    auto mu_part = torch::empty({b, rchunks}, xq.options());
    auto dist_part = torch::empty({b, rchunks}, xq.options());
    launch_ps_kvec_split(
        xq.data_ptr<float>(), x.data_ptr<float>(),
        inv_ls.data_ptr<float>(), alpha.data_ptr<float>(),
        onehot.data_ptr<unsigned char>(), k_ws.data_ptr<float>(),
        mu_part.data_ptr<float>(), dist_part.data_ptr<float>(),
        mu_ws.data_ptr<float>(), dist_ws.data_ptr<float>(), b, n, d,
        amp2, rchunks, current_stream());
    auto quad = quadform_large_n(k_ws, kinv, b, n, current_stream());
    launch_ps_finalize_meanstd_direct(
        mu_ws.data_ptr<float>(), quad.data_ptr<float>(),
        mean.data_ptr<float>(), sd.data_ptr<float>(), b, amp2,
        (float)mean_c, current_stream());
    return {mean, sd, dist_ws};
  }
  launch_ps_kvec(
      xq.data_ptr<float>(), x.data_ptr<float>(),
      inv_ls.data_ptr<float>(), alpha.data_ptr<float>(),
      onehot.data_ptr<unsigned char>(), k_ws.data_ptr<float>(),
      mu_ws.data_ptr<float>(), dist_ws.data_ptr<float>(), b, n, d,
      amp2, current_stream());
  if (b <= 32) {
    auto quad = quadform_tile_small_n(k_ws, kinv, b, n,
                                      current_stream());
    launch_ps_finalize_meanstd_direct(
        mu_ws.data_ptr<float>(), quad.data_ptr<float>(),
        mean.data_ptr<float>(), sd.data_ptr<float>(), b, amp2,
        (float)mean_c, current_stream());
    return {mean, sd, dist_ws};
  }
  auto var_ws = torch::empty({b, 10}, xq.options());
  launch_ps_quadform_kernel_only(
      k_ws.data_ptr<float>(), kinv.data_ptr<float>(),
      var_ws.data_ptr<float>(), b, n, current_stream());
  launch_ps_finalize_meanstd(
      mu_ws.data_ptr<float>(), var_ws.data_ptr<float>(),
      mean.data_ptr<float>(), sd.data_ptr<float>(), b, amp2,
      (float)mean_c, 10, current_stream());
  return {mean, sd, dist_ws};
}

std::vector<torch::Tensor> posterior_mean_std_fp8(
    torch::Tensor xq, torch::Tensor x, torch::Tensor z2q,
    torch::Tensor n2, double scale, torch::Tensor lengthscales,
    double amplitude, double mean_c, torch::Tensor alpha,
    torch::Tensor kinv, torch::Tensor onehot) {
  // (mean, sd, dist) with fp8 candidate grams against cached training
  // operands (Fp8GramCache layout) — the per-metric piece of the
  // fused MO fp8 scorer (config 5).
  xq = check_f32(xq, "xq");
  x = check_f32(x, "x");
  n2 = check_f32(n2, "n2");
  lengthscales = check_f32(lengthscales, "lengthscales");
  alpha = check_f32(alpha, "alpha");
  kinv = check_f32(kinv, "kinv");
  TORCH_CHECK(z2q.scalar_type() == torch::kFloat8_e4m3fn,
              "z2q must be float8_e4m3fn");
  z2q = z2q.contiguous();
  onehot = onehot.contiguous();
  const int b = xq.size(0), d = xq.size(1), n = x.size(0);
  const int dp = z2q.size(1);
  TORCH_CHECK(dp % 32 == 0 && dp <= 512, "dp must be mult of 32, <=512");
  // Candidate side: quantize + dequantize with torch (tiny, no sync).
  auto z1 = xq / lengthscales;
  auto z1q = torch::zeros({b, dp},
                          xq.options().dtype(torch::kFloat8_e4m3fn));
  z1q.index_put_({torch::indexing::Slice(),
                  torch::indexing::Slice(0, d)},
                 (z1 / scale).to(torch::kFloat8_e4m3fn));
  auto z1f = z1q.to(torch::kFloat32) * scale;
  auto n1 = (z1f * z1f).sum(-1);
  auto k_ws = torch::empty({b, n}, xq.options());
  auto mu_ws = torch::empty({b}, xq.options());
  auto dist_ws = torch::empty({b}, xq.options());
  auto mean = torch::empty({b}, xq.options());
  auto sd = torch::empty({b}, xq.options());
  const float amp2 = (float)(amplitude * amplitude);
  launch_ps_kvec_fp8(
      z1f.contiguous().data_ptr<float>(), n1.contiguous().data_ptr<float>(),
      xq.data_ptr<float>(), x.data_ptr<float>(),
      (const unsigned char*)z2q.data_ptr(), n2.data_ptr<float>(),
      alpha.data_ptr<float>(), onehot.data_ptr<unsigned char>(),
      k_ws.data_ptr<float>(), mu_ws.data_ptr<float>(),
      dist_ws.data_ptr<float>(), b, n, d, dp, amp2, (float)scale,
      current_stream());
  static const int gemm_n_threshold = []() {
    const char* sE = getenv("VIZIER_AMD_PS_GEMM_N");
    return sE ? atoi(sE) : 4096;
  }();
  auto quad = (n < gemm_n_threshold && b <= 32)
      ? quadform_tile_small_n(k_ws, kinv, b, n, current_stream())
      : quadform_large_n(k_ws, kinv, b, n, current_stream());
  launch_ps_finalize_meanstd_direct(
      mu_ws.data_ptr<float>(), quad.data_ptr<float>(),
      mean.data_ptr<float>(), sd.data_ptr<float>(), b, amp2,
      (float)mean_c, current_stream());
  return {mean, sd, dist_ws};
}

torch::Tensor hv_scalarize_tr(
    torch::Tensor means, torch::Tensor sds, torch::Tensor weights,
    c10::optional<torch::Tensor> ref, c10::optional<torch::Tensor> dist,
    double coef, double tr_radius) {
  means = check_f32(means, "means");    // (M, B)
  sds = check_f32(sds, "sds");
  weights = check_f32(weights, "weights");  // (S, M)
  const int m = means.size(0), b = means.size(1);
  const int s = weights.size(0);
  TORCH_CHECK(weights.size(1) == m, "weights/means metric mismatch");
  torch::Tensor ref_t, dist_t;
  const float* ref_p = nullptr;
  const float* dist_p = nullptr;
  if (ref.has_value()) {
    ref_t = check_f32(*ref, "ref");
    ref_p = ref_t.data_ptr<float>();
  }
  if (dist.has_value()) {
    dist_t = check_f32(*dist, "dist");
    dist_p = dist_t.data_ptr<float>();
  }
  auto out = torch::empty({b}, means.options());
  launch_hv_scalarize_tr(
      means.data_ptr<float>(), sds.data_ptr<float>(),
      weights.data_ptr<float>(), ref_p, dist_p,
      out.data_ptr<float>(), b, m, s, (float)coef, (float)tr_radius,
      current_stream());
  return out;
}

std::vector<torch::Tensor> batched_potrf(torch::Tensor K) {
  // (R, N, N) -> (L lower in a fresh tensor, info (R,) int32).
  K = check_f32(K, "K");
  TORCH_CHECK(K.dim() == 3 && K.size(1) == K.size(2),
              "K must be (R, N, N)");
  const int r = K.size(0), n = K.size(1);
  auto info = torch::zeros({r}, K.options().dtype(torch::kInt32));
  // Implementation ladder (VIZIER_AMD_CHOL_IMPL=v2|v3|v4|v5, default
  // v2 — every alternative was A/B'd SLOWER at the headline shape):
  //   v2 (default): 2 launches per 32-panel round; 2.85 ms at
  //       (R=3..12, N=1000). Execution-bound: a 63-deep trivial-kernel
  //       chain runs at 4.4 us/launch, LDS-staged coalescing changed
  //       nothing, so the per-kernel ~45 us is genuine work+fence.
  //   v5: persistent left-looking cooperative kernel (read-only input
  //       stays cached, panel-major once-written output, diagonal via
  //       agent-scope scratch) — 3.15 ms: the per-panel LDS staging
  //       loop (~500 sync'd iterations) and occupancy 2 eat the
  //       saved inter-kernel fences.
  //   v3: right-looking cooperative — 2x slower (trailing matrix is
  //       cross-workgroup mutable -> all traffic memory-side).
  //   v4: fused one-kernel-per-round with recompute — occupancy 1
  //       (VGPR 256 + 248 AGPRs).
  static const int impl = []() {
    const char* env = std::getenv("VIZIER_AMD_CHOL_IMPL");
    if (env == nullptr) return 2;
    const std::string s(env);
    if (s == "v5") return 5;
    if (s == "v3") return 3;
    return 2;  // v2/v4 re-checked downstream
  }();
  if (impl == 5) {
    auto L = torch::empty_like(K);
    const long pk = (n + 31) / 32;
    auto panels = torch::empty({(long)r * pk * n * 32}, K.options());
    auto dscratch = torch::empty({(long)r * 32 * 32}, K.options());
    auto bar = torch::zeros({2}, K.options().dtype(torch::kInt32));
    if (launch_batched_potrf_v5(
            K.data_ptr<float>(), panels.data_ptr<float>(),
            dscratch.data_ptr<float>(), L.data_ptr<float>(),
            info.data_ptr<int>(),
            reinterpret_cast<unsigned int*>(bar.data_ptr<int>()),
            r, n, current_stream()) == 0) {
      return {L, info};
    }
  }
  auto L = K.clone();
  if (impl == 3) {
    auto bar = torch::zeros({2}, K.options().dtype(torch::kInt32));
    if (launch_batched_potrf_coop(
            L.data_ptr<float>(), info.data_ptr<int>(),
            reinterpret_cast<unsigned int*>(bar.data_ptr<int>()),
            r, n, current_stream()) == 0) {
      return {L, info};
    }
  }
  launch_batched_potrf(L.data_ptr<float>(), info.data_ptr<int>(), r, n,
                       current_stream());
  return {L, info};
}

torch::Tensor batched_trsv_lower(torch::Tensor L, torch::Tensor b) {
  L = check_f32(L, "L");
  b = check_f32(b, "b");
  TORCH_CHECK(L.dim() == 3 && L.size(1) == L.size(2) &&
              b.dim() == 2 && b.size(0) == L.size(0) &&
              b.size(1) == L.size(1), "shape mismatch");
  auto z = b.clone();
  launch_batched_trsv_lower(L.data_ptr<float>(), z.data_ptr<float>(),
                            (int)L.size(0), (int)L.size(1),
                            current_stream());
  return z;
}

std::vector<torch::Tensor> eagle_suggest(
    torch::Tensor pool_cont, torch::Tensor pool_cat, torch::Tensor rewards,
    torch::Tensor perturbations, torch::Tensor cat_sizes,
    torch::Tensor iter_counter, int64_t n_batches, int64_t batch_size,
    double visibility, double gravity, double neg_gravity,
    double norm_scale, double cat_factor, double p_same, int64_t seed,
    torch::Tensor out_cont, torch::Tensor out_cat, int64_t max_cat) {
  pool_cont = check_f32(pool_cont, "pool_cont");
  rewards = check_f32(rewards, "rewards");
  perturbations = check_f32(perturbations, "perturbations");
  TORCH_CHECK(iter_counter.scalar_type() == torch::kInt64 &&
              iter_counter.is_cuda(), "iter_counter must be int64 cuda");
  pool_cat = pool_cat.contiguous();
  cat_sizes = cat_sizes.contiguous();
  const int pool_size = pool_cont.size(0);
  const int q = pool_cont.size(1);
  const int dc = pool_cont.size(2);
  const int dcat = pool_cat.size(2);
  launch_eagle_suggest(
      pool_cont.data_ptr<float>(),
      dcat ? pool_cat.data_ptr<long>() : nullptr,
      rewards.data_ptr<float>(), perturbations.data_ptr<float>(),
      dcat ? cat_sizes.data_ptr<long>() : nullptr,
      out_cont.data_ptr<float>(),
      dcat ? out_cat.data_ptr<long>() : nullptr,
      (const unsigned long long*)iter_counter.data_ptr<int64_t>(),
      (int)n_batches, (int)batch_size, pool_size, q, dc, dcat,
      (int)max_cat, (float)visibility,
      (float)gravity, (float)neg_gravity, (float)norm_scale,
      (float)cat_factor, (float)p_same, (unsigned long long)seed,
      current_stream());
  return {out_cont, out_cat};
}

void eagle_update(torch::Tensor pool_cont, torch::Tensor pool_cat,
                  torch::Tensor rewards, torch::Tensor perturbations,
                  torch::Tensor batch_cont, torch::Tensor batch_cat,
                  torch::Tensor batch_rewards, torch::Tensor cat_sizes,
                  torch::Tensor best_reward, torch::Tensor iter_counter,
                  int64_t n_batches,
                  double penalize_factor, double perturbation_lower_bound,
                  double base_perturbation, int64_t seed) {
  const int batch_size = batch_cont.size(0);
  const int q = batch_cont.size(1);
  const int dc = batch_cont.size(2);
  const int dcat = batch_cat.size(2);
  launch_eagle_update(
      pool_cont.data_ptr<float>(),
      dcat ? pool_cat.data_ptr<long>() : nullptr,
      rewards.data_ptr<float>(), perturbations.data_ptr<float>(),
      batch_cont.data_ptr<float>(),
      dcat ? batch_cat.data_ptr<long>() : nullptr,
      batch_rewards.data_ptr<float>(),
      dcat ? cat_sizes.data_ptr<long>() : nullptr,
      best_reward.data_ptr<float>(),
      (unsigned long long*)iter_counter.data_ptr<int64_t>(),
      (int)n_batches, batch_size, q, dc,
      dcat, (float)penalize_factor, (float)perturbation_lower_bound,
      (float)base_perturbation, (unsigned long long)seed,
      current_stream());
}

int64_t eagle_sweep(
    torch::Tensor pool_cont, torch::Tensor rewards,
    torch::Tensor perturbations, torch::Tensor best_reward,
    torch::Tensor iter_counter, torch::Tensor barrier_buf,
    torch::Tensor x, torch::Tensor inv_ls,
    torch::Tensor alpha, torch::Tensor kinv, torch::Tensor out_cont,
    torch::Tensor k_ws, torch::Tensor mu_ws, torch::Tensor dist_ws,
    torch::Tensor var_ws, int64_t n_batches,
    int64_t batch_size, int64_t pool_size, int64_t it_start,
    int64_t iterations, double visibility, double gravity,
    double neg_gravity, double norm_scale, double penalize_factor,
    double perturbation_lower_bound, double base_perturbation,
    int64_t seed_suggest, int64_t seed_update, double amp2,
    double mean_c, int64_t acq, double coef, double best_value,
    double tr_radius) {
  pool_cont = check_f32(pool_cont, "pool_cont");
  rewards = check_f32(rewards, "rewards");
  perturbations = check_f32(perturbations, "perturbations");
  x = check_f32(x, "x");
  inv_ls = check_f32(inv_ls, "inv_ls");
  alpha = check_f32(alpha, "alpha");
  kinv = check_f32(kinv, "kinv");
  TORCH_CHECK(barrier_buf.scalar_type() == torch::kInt32 &&
              barrier_buf.numel() >= 2, "barrier_buf must be int32 (2,)");
  const int dc = x.size(1);
  const int n = x.size(0);
  TORCH_CHECK(pool_cont.numel() == pool_size * dc,
              "pool/feature shape mismatch (continuous-only, q=1)");
  TORCH_CHECK(pool_size <= 128, "eagle_sweep supports pool <= 128");
  TORCH_CHECK(n <= 8192, "eagle_sweep supports N <= 8192");
  TORCH_CHECK(batch_size <= 32,
              "eagle_sweep tile quadform supports batch <= 32");
  const long tiles_n = (n + 63) / 64;
  TORCH_CHECK(var_ws.numel() >= batch_size * (tiles_n * tiles_n + 1),
              "var_ws must be (B, tiles^2 + 1)");
  const int ret = launch_eagle_sweep(
      pool_cont.data_ptr<float>(), rewards.data_ptr<float>(),
      perturbations.data_ptr<float>(), best_reward.data_ptr<float>(),
      (unsigned long long*)iter_counter.data_ptr<int64_t>(),
      (unsigned int*)barrier_buf.data_ptr<int>(),
      x.data_ptr<float>(), inv_ls.data_ptr<float>(),
      alpha.data_ptr<float>(), kinv.data_ptr<float>(),
      out_cont.data_ptr<float>(), k_ws.data_ptr<float>(),
      mu_ws.data_ptr<float>(), dist_ws.data_ptr<float>(),
      var_ws.data_ptr<float>(),
      (int)n_batches, (int)batch_size, (int)pool_size, dc, n,
      (long long)it_start, (long long)iterations, (float)visibility,
      (float)gravity, (float)neg_gravity, (float)norm_scale,
      (float)penalize_factor, (float)perturbation_lower_bound,
      (float)base_perturbation, (unsigned long long)seed_suggest,
      (unsigned long long)seed_update, (float)amp2, (float)mean_c,
      (int)acq, (float)coef, (float)best_value, (float)tr_radius,
      current_stream());
  TORCH_CHECK(ret > 0, "cooperative eagle_sweep launch failed (code ",
              ret, ")");
  return ret;
}


}  // namespace

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
  m.def("gram_matern52_batched", &gram_matern52_batched,
        "Batched-restart fused Matern-5/2 gram (+noise*I, optional "
        "gradient factor)");
  m.def("gram_matern52", &gram_matern52,
        "Fused Matern-5/2 ARD Gram matrix (gfx950)");
  m.def("gram_matern52_bf16", &gram_matern52_bf16,
        "bf16 MFMA Matern-5/2 Gram matrix (gfx950 matrix cores)");
  m.def("gram_matern52_bf16_tiled", &gram_matern52_bf16_tiled,
        "128x128 LDS-tiled bf16 MFMA Matern-5/2 Gram (gfx950)");
  m.def("gram_matern52_fp8", &gram_matern52_fp8,
        "fp8 e4m3 MFMA Matern-5/2 Gram matrix (gfx950, config 5)");
  m.def("gram_matern52_fp8_tiled", &gram_matern52_fp8_tiled,
        "128x128 LDS-tiled fp8 e4m3 MFMA Matern-5/2 Gram (gfx950)");
  m.def("posterior_scores", &posterior_scores,
        "Fused GP posterior + acquisition + trust region (gfx950)");
  m.def("batched_potrf", &batched_potrf,
        "Batched lower Cholesky, one workgroup per matrix (gfx950)");
  m.def("batched_trsv_lower", &batched_trsv_lower,
        "Batched forward substitution L z = b (gfx950)");
  m.def("gram_matern52_fp8_pre", &gram_matern52_fp8_pre,
        "fp8 cross-gram from pre-quantized operands (gfx950)");
  m.def("posterior_mean_std_fp8", &posterior_mean_std_fp8,
        "Fused (mean, sd, dist) with fp8 cached-operand grams");
  m.def("posterior_mean_std", &posterior_mean_std,
        "Fused (mean, sd, dist) for one GP over candidates (gfx950)");
  m.def("hv_scalarize_tr", &hv_scalarize_tr,
        "Hypervolume scalarization + trust region in one launch");
  m.def("posterior_scores_bf16", &posterior_scores_bf16,
        "Chunked scorer with cached-bf16 candidate grams (gfx950)");
  m.def("posterior_scores_chunked", &posterior_scores_chunked,
        "3-kernel chunked GP posterior scorer (chip-filling, gfx950)");
  m.def("eagle_suggest", &eagle_suggest,
        "Fused Eagle suggest step (gfx950)");
  m.def("eagle_update", &eagle_update, "Fused Eagle update step (gfx950)");
  m.def("eagle_sweep", &eagle_sweep,
        "Persistent cooperative Eagle sweep megakernel (gfx950)");
}
