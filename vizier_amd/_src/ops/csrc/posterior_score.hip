// Fused GP-posterior + acquisition + trust-region kernel for gfx950.
//
// For each candidate q (one 256-thread workgroup per candidate):
//   k[n]   = amp^2 * m52(||(xq - x_n)/ls||)          (n = 0..N-1)
//   mu     = mean + sum_n k[n] * alpha[n]
//   var    = amp^2 - k^T Kinv k        (Kinv = (K + noise I)^-1, symmetric)
//   dist   = min_n max_d |xq_d - x_nd| (continuous dims only)
//   score  = acq(mu, sqrt(var)) or -1e4 - dist outside the trust region
//
// This replaces ~15 separate eager launches per Eagle iteration with one
// launch, and turns the per-candidate triangular solve of the reference
// (latency-bound) into a quadratic form against the precomputed Kinv
// (GEMM-shaped, L2/L3-resident: Kinv is reused across all 3000
// iterations of a sweep). Numeric spec: GPPosterior.predict +
// ScoringFunction in vizier_amd/_src/gp.
//
// k[] is staged in LDS (cap N_LDS rows; larger N falls back to a
// strided two-pass global path handled by the host).

#include <hip/hip_runtime.h>
#include "common.h"

#define BLOCK 256
#define MAX_N_LDS 8192  // 32 KiB of k[] in LDS

// acquisition codes
#define ACQ_UCB 0
#define ACQ_LCB 1
#define ACQ_EI 2
#define ACQ_PI 3
#define ACQ_MEAN 4
#define ACQ_STDDEV 5

__device__ __forceinline__ float normal_cdf(float z) {
  return 0.5f * erfcf(-z * 0.70710678118654752f);
}
__device__ __forceinline__ float normal_pdf(float z) {
  return 0.3989422804014327f * __expf(-0.5f * z * z);
}

extern "C" __global__ __launch_bounds__(BLOCK) void
posterior_score_kernel(const float* __restrict__ xq,     // (B, D)
                       const float* __restrict__ x,      // (N, D)
                       const float* __restrict__ inv_ls, // (D,)
                       const float* __restrict__ alpha,  // (N,)
                       const float* __restrict__ kinv,   // (N, N)
                       const unsigned char* __restrict__ onehot,  // (D,)
                       float* __restrict__ out,          // (B,)
                       int b, int n, int d, float amp2, float mean_c,
                       int acq, float coef, float best_value,
                       float tr_radius) {
  extern __shared__ float lds[];        // k[n] (n <= MAX_N_LDS)
  __shared__ float red[8];
  __shared__ float xq_lds[512];         // candidate features (D <= 512)

  const int q = blockIdx.x;
  if (q >= b) return;
  const int tid = threadIdx.x;

  for (int j = tid; j < d; j += BLOCK) xq_lds[j] = xq[q * d + j];
  __syncthreads();

  // Phase 1: k-vector, mu partial, trust-region distance.
  float mu_acc = 0.0f;
  float min_linf = INFINITY;
  for (int row = tid; row < n; row += BLOCK) {
    const float* xr = x + row * d;
    float d2 = 0.0f;
    float linf = 0.0f;
    for (int j = 0; j < d; ++j) {
      const float diff = xq_lds[j] - xr[j];
      const float z = diff * inv_ls[j];
      d2 = fmaf(z, z, d2);
      if (!onehot[j]) linf = fmaxf(linf, fabsf(diff));
    }
    const float kv = amp2 * matern52_of_d2(d2);
    lds[row] = kv;
    mu_acc = fmaf(kv, alpha[row], mu_acc);
    min_linf = fminf(min_linf, linf);
  }
  __syncthreads();

  // Phase 2: quadform var = sum_j k[j] * (sum_i kinv[j,i] k[i]).
  // Thread t owns rows j = t, t+BLOCK, ...; row walk is contiguous.
  float var_acc = 0.0f;
  for (int j = tid; j < n; j += BLOCK) {
    const float kj = lds[j];
    const float* row = kinv + (long)j * n;
    float t_j = 0.0f;
    int i = 0;
    for (; i + 4 <= n; i += 4) {
      t_j = fmaf(row[i], lds[i], t_j);
      t_j = fmaf(row[i + 1], lds[i + 1], t_j);
      t_j = fmaf(row[i + 2], lds[i + 2], t_j);
      t_j = fmaf(row[i + 3], lds[i + 3], t_j);
    }
    for (; i < n; ++i) t_j = fmaf(row[i], lds[i], t_j);
    var_acc = fmaf(kj, t_j, var_acc);
  }

  // Block reductions.
  auto fsum = [](float a, float c) { return a + c; };
  auto fmin_ = [](float a, float c) { return fminf(a, c); };
  float mu = block_reduce(mu_acc, red, fsum, 0.0f);
  __syncthreads();
  float var = block_reduce(var_acc, red, fsum, 0.0f);
  __syncthreads();
  float dist = block_reduce(min_linf, red, fmin_, INFINITY);

  if (tid == 0) {
    mu += mean_c;
    var = fmaxf(amp2 - var, 1e-12f);
    const float sd = sqrtf(var);
    float score;
    switch (acq) {
      case ACQ_LCB: score = mu - coef * sd; break;
      case ACQ_EI: {
        const float z = (mu - best_value) / sd;
        score = sd * (z * normal_cdf(z) + normal_pdf(z));
        break;
      }
      case ACQ_PI: {
        const float z = (mu - best_value) / sd;
        score = normal_cdf(z);
        break;
      }
      case ACQ_MEAN: score = mu; break;
      case ACQ_STDDEV: score = sd; break;
      case ACQ_UCB:
      default: score = mu + coef * sd; break;
    }
    if (tr_radius > 0.0f && tr_radius <= 0.5f && dist > tr_radius) {
      score = -1e4f - dist;
    }
    out[q] = score;
  }
}

extern "C" void launch_posterior_score(
    const float* xq, const float* x, const float* inv_ls,
    const float* alpha, const float* kinv, const unsigned char* onehot,
    float* out, int b, int n, int d, float amp2, float mean_c, int acq,
    float coef, float best_value, float tr_radius, hipStream_t stream) {
  dim3 grid(b);
  dim3 block(BLOCK);
  size_t shmem = (size_t)n * sizeof(float);
  hipLaunchKernelGGL(posterior_score_kernel, grid, block, shmem, stream,
                     xq, x, inv_ls, alpha, kinv, onehot, out, b, n, d,
                     amp2, mean_c, acq, coef, best_value, tr_radius);
}

// -- Chunked 3-kernel variant -------------------------------------------
//
// The single-workgroup-per-candidate kernel above underfills the chip at
// small batch sizes (B=25 -> 25 of 256 CUs). This variant splits the
// K^-1 quadform over NCHUNK j-ranges so B x NCHUNK workgroups run the
// heavy phase; the three kernels are launched back-to-back from one
// binding call (still a single Python-level op, hipGraph-capturable).

#define NCHUNK 10

extern "C" __global__ __launch_bounds__(BLOCK) void
ps_kvec_kernel(const float* __restrict__ xq, const float* __restrict__ x,
               const float* __restrict__ inv_ls,
               const float* __restrict__ alpha,
               const unsigned char* __restrict__ onehot,
               float* __restrict__ k_out,    // (B, N)
               float* __restrict__ mu_out,   // (B,)
               float* __restrict__ dist_out, // (B,)
               int b, int n, int d, float amp2) {
  __shared__ float red[8];
  __shared__ float xq_lds[512];
  const int q = blockIdx.x;
  if (q >= b) return;
  const int tid = threadIdx.x;
  for (int j = tid; j < d; j += BLOCK) xq_lds[j] = xq[q * d + j];
  __syncthreads();
  float mu_acc = 0.0f;
  float min_linf = INFINITY;
  for (int row = tid; row < n; row += BLOCK) {
    const float* xr = x + (long)row * d;
    float d2 = 0.0f, linf = 0.0f;
    for (int j = 0; j < d; ++j) {
      const float diff = xq_lds[j] - xr[j];
      const float z = diff * inv_ls[j];
      d2 = fmaf(z, z, d2);
      if (!onehot[j]) linf = fmaxf(linf, fabsf(diff));
    }
    const float kv = amp2 * matern52_of_d2(d2);
    k_out[(long)q * n + row] = kv;
    mu_acc = fmaf(kv, alpha[row], mu_acc);
    min_linf = fminf(min_linf, linf);
  }
  auto fsum = [](float a, float c) { return a + c; };
  auto fmin_ = [](float a, float c) { return fminf(a, c); };
  float mu = block_reduce(mu_acc, red, fsum, 0.0f);
  if (tid == 0) mu_out[q] = mu;
  __syncthreads();
  float dist = block_reduce(min_linf, red, fmin_, INFINITY);
  if (tid == 0) dist_out[q] = dist;
}

extern "C" __global__ __launch_bounds__(BLOCK) void
ps_quadform_kernel(const float* __restrict__ k_in,   // (B, N)
                   const float* __restrict__ kinv,   // (N, N)
                   float* __restrict__ var_part,     // (B, NCHUNK)
                   int b, int n) {
  // Wave-per-row: all 64 lanes of a wave stream ONE Kinv row with
  // coalesced float4 loads and shuffle-reduce the dot with k; a
  // per-thread row walk here was 64 divergent streams (measured 65us,
  // ~60 GB/s effective — see profiles/bench_kernel_stats.csv).
  __shared__ float red[8];
  const int q = blockIdx.x;
  const int chunk = blockIdx.y;
  const int tid = threadIdx.x;
  const int wave = tid / WAVE_SIZE;
  const int lane = tid % WAVE_SIZE;
  const int waves = BLOCK / WAVE_SIZE;
  const int j0 = (int)((long)chunk * n / NCHUNK);
  const int j1 = (int)((long)(chunk + 1) * n / NCHUNK);
  const float* k = k_in + (long)q * n;
  float acc = 0.0f;
  const int n4 = n / 4;
  // float4 loads are only legal when every row base (kinv + j*n, k_in +
  // q*n) is 16-byte aligned, i.e. n % 4 == 0 (the host pads the
  // workspace leading dim, but defend here too: misaligned vector
  // loads are UB on gfx950).
  if ((n & 3) == 0) {
    for (int j = j0 + wave; j < j1; j += waves) {
      const float4* row4 = reinterpret_cast<const float4*>(
          kinv + (long)j * n);
      const float4* k4 = reinterpret_cast<const float4*>(k);
      float t_j = 0.0f;
      for (int i4 = lane; i4 < n4; i4 += WAVE_SIZE) {
        const float4 r = row4[i4];
        const float4 kv = k4[i4];
        t_j = fmaf(r.x, kv.x, t_j);
        t_j = fmaf(r.y, kv.y, t_j);
        t_j = fmaf(r.z, kv.z, t_j);
        t_j = fmaf(r.w, kv.w, t_j);
      }
      t_j = wave_reduce_sum(t_j);
      if (lane == 0) acc = fmaf(k[j], t_j, acc);
    }
  } else {
    for (int j = j0 + wave; j < j1; j += waves) {
      const float* row = kinv + (long)j * n;
      float t_j = 0.0f;
      for (int i = lane; i < n; i += WAVE_SIZE) {
        t_j = fmaf(row[i], k[i], t_j);
      }
      t_j = wave_reduce_sum(t_j);
      if (lane == 0) acc = fmaf(k[j], t_j, acc);
    }
  }
  // acc lives in lane 0 of each wave; combine across waves.
  auto fsum = [](float a, float c) { return a + c; };
  float v = block_reduce(acc, red, fsum, 0.0f);
  if (tid == 0) var_part[q * NCHUNK + chunk] = v;
}

extern "C" __global__ void
ps_finalize_kernel(const float* __restrict__ mu_in,
                   const float* __restrict__ dist_in,
                   const float* __restrict__ var_part,
                   float* __restrict__ out, int b, float amp2,
                   float mean_c, int acq, float coef, float best_value,
                   float tr_radius) {
  const int q = blockIdx.x * blockDim.x + threadIdx.x;
  if (q >= b) return;
  float var = 0.0f;
  for (int c = 0; c < NCHUNK; ++c) var += var_part[q * NCHUNK + c];
  var = fmaxf(amp2 - var, 1e-12f);
  const float sd = sqrtf(var);
  const float mu = mu_in[q] + mean_c;
  float score;
  switch (acq) {
    case ACQ_LCB: score = mu - coef * sd; break;
    case ACQ_EI: {
      const float z = (mu - best_value) / sd;
      score = sd * (z * normal_cdf(z) + normal_pdf(z));
      break;
    }
    case ACQ_PI: {
      const float z = (mu - best_value) / sd;
      score = normal_cdf(z);
      break;
    }
    case ACQ_MEAN: score = mu; break;
    case ACQ_STDDEV: score = sd; break;
    case ACQ_UCB:
    default: score = mu + coef * sd; break;
  }
  const float dist = dist_in[q];
  if (tr_radius > 0.0f && tr_radius <= 0.5f && dist > tr_radius) {
    score = -1e4f - dist;
  }
  out[q] = score;
}

// -- bf16 k-vector with CACHED training operands ------------------------
//
// Config-2 parity (BASELINE bf16): round 1 ran bf16 candidate grams on
// the composed eager path, paying per-call conversion of the TRAINING
// features (N x D host-side ops, ~15 launches/iteration: 814 ms vs
// 197 ms fp32 per suggest). Here the training side (z2b = x/ls rounded
// to bf16, n2 = rounded row norms) is converted ONCE per suggest by the
// Python caller and reused across all 3000 sweep iterations; this
// kernel only rounds the 25 candidates, making the whole bf16 scorer
// a 3-launch graph-capturable sequence like the fp32 path.
// d^2 = n1 + n2 - 2 z1.z2 with all products of bf16-rounded values
// accumulated in fp32 — the exact squared distance of the rounded
// inputs (same numerics as gram_matern52_bf16).

__device__ __forceinline__ unsigned short vz_f2bf(float f) {
  // Round-to-nearest-even, matching torch's .to(bfloat16).
  unsigned int u = __float_as_uint(f);
  u += 0x7fffu + ((u >> 16) & 1u);
  return (unsigned short)(u >> 16);
}
__device__ __forceinline__ float vz_bf2f(unsigned short h) {
  return __uint_as_float(((unsigned int)h) << 16);
}

extern "C" __global__ __launch_bounds__(BLOCK) void
ps_kvec_bf16_kernel(const float* __restrict__ xq,        // (B, D)
                    const float* __restrict__ x,         // (N, D) raw
                    const unsigned short* __restrict__ z2b,  // (N, Dp)
                    const float* __restrict__ n2,        // (N,)
                    const float* __restrict__ inv_ls,    // (D,)
                    const float* __restrict__ alpha,     // (N,)
                    const unsigned char* __restrict__ onehot,
                    float* __restrict__ k_out,           // (B, N)
                    float* __restrict__ mu_out,          // (B,)
                    float* __restrict__ dist_out,        // (B,)
                    int b, int n, int d, int dp, float amp2) {
  __shared__ float red[8];
  __shared__ float xq_lds[512];       // raw candidate (trust distance)
  __shared__ float z1_lds[512];       // bf16-ROUNDED scaled candidate
  __shared__ float n1_sh;
  const int q = blockIdx.x;
  if (q >= b) return;
  const int tid = threadIdx.x;
  float n1_acc = 0.0f;
  for (int j = tid; j < dp; j += BLOCK) {
    float raw = 0.0f, v = 0.0f;
    if (j < d) {
      raw = xq[(long)q * d + j];
      v = vz_bf2f(vz_f2bf(raw * inv_ls[j]));
      xq_lds[j] = raw;
    }
    z1_lds[j] = v;
    n1_acc = fmaf(v, v, n1_acc);
  }
  __syncthreads();
  auto fsum = [](float a, float c) { return a + c; };
  float n1 = block_reduce(n1_acc, red, fsum, 0.0f);
  if (tid == 0) n1_sh = n1;
  __syncthreads();
  n1 = n1_sh;

  float mu_acc = 0.0f;
  float min_linf = INFINITY;
  for (int row = tid; row < n; row += BLOCK) {
    const unsigned short* zr = z2b + (long)row * dp;
    float dot = 0.0f;
    for (int j = 0; j < dp; ++j) {
      dot = fmaf(z1_lds[j], vz_bf2f(zr[j]), dot);
    }
    const float d2 = fmaxf(n1 + n2[row] - 2.0f * dot, 0.0f);
    const float kv = amp2 * matern52_of_d2(d2);
    k_out[(long)q * n + row] = kv;
    mu_acc = fmaf(kv, alpha[row], mu_acc);
    const float* xr = x + (long)row * d;
    float linf = 0.0f;
    for (int j = 0; j < d; ++j) {
      if (!onehot[j]) linf = fmaxf(linf, fabsf(xq_lds[j] - xr[j]));
    }
    min_linf = fminf(min_linf, linf);
  }
  auto fmin_ = [](float a, float c) { return fminf(a, c); };
  float mu = block_reduce(mu_acc, red, fsum, 0.0f);
  if (tid == 0) mu_out[q] = mu;
  __syncthreads();
  float dist = block_reduce(min_linf, red, fmin_, INFINITY);
  if (tid == 0) dist_out[q] = dist;
}

extern "C" void launch_ps_kvec_bf16(
    const float* xq, const float* x, const unsigned short* z2b,
    const float* n2, const float* inv_ls, const float* alpha,
    const unsigned char* onehot, float* k_ws, float* mu_ws,
    float* dist_ws, int b, int n, int d, int dp, float amp2,
    hipStream_t stream) {
  hipLaunchKernelGGL(ps_kvec_bf16_kernel, dim3(b), dim3(BLOCK), 0, stream,
                     xq, x, z2b, n2, inv_ls, alpha, onehot, k_ws, mu_ws,
                     dist_ws, b, n, d, dp, amp2);
}

// -- fp8 (e4m3fn) k-vector with cached training operands ----------------
//
// Same shape as the bf16 variant: the training side is quantized ONCE
// per suggest (Fp8GramCache: z2q = fp8(x/ls/s), n2 = dequantized row
// norms, s = unit-box bound so no per-call range scan). The CANDIDATE
// side arrives pre-dequantized from the binding (4 tiny torch ops,
// capture-safe); this kernel software-decodes the cached e4m3fn bytes
// and computes d^2 = n1 + n2 - 2 s * z1f . z2 exactly as the fp8 MFMA
// gram does (gram_matern52_fp8.hip).

__device__ __forceinline__ float vz_fp8_to_f32(unsigned char v) {
  const int sign = v >> 7;
  const int e = (v >> 3) & 0xF;
  const int m = v & 7;
  float f;
  if (e == 0) {
    f = ldexpf((float)m, -9);              // subnormal: m * 2^-9
  } else {
    f = ldexpf(1.0f + m * 0.125f, e - 7);  // bias 7
  }
  return sign ? -f : f;
}

extern "C" __global__ __launch_bounds__(BLOCK) void
ps_kvec_fp8_kernel(const float* __restrict__ z1f,   // (B, Dp) dequant
                   const float* __restrict__ n1,    // (B,)
                   const float* __restrict__ xq,    // (B, D) raw
                   const float* __restrict__ x,     // (N, D) raw
                   const unsigned char* __restrict__ z2q,  // (N, Dp)
                   const float* __restrict__ n2,    // (N,)
                   const float* __restrict__ alpha, // (N,)
                   const unsigned char* __restrict__ onehot,
                   float* __restrict__ k_out, float* __restrict__ mu_out,
                   float* __restrict__ dist_out, int b, int n, int d,
                   int dp, float amp2, float scale) {
  __shared__ float red[8];
  __shared__ float xq_lds[512];
  __shared__ float z1_lds[512];
  const int q = blockIdx.x;
  if (q >= b) return;
  const int tid = threadIdx.x;
  for (int j = tid; j < dp; j += BLOCK) {
    z1_lds[j] = z1f[(long)q * dp + j];
    if (j < d) xq_lds[j] = xq[(long)q * d + j];
  }
  __syncthreads();
  const float n1q = n1[q];
  float mu_acc = 0.0f;
  float min_linf = INFINITY;
  for (int row = tid; row < n; row += BLOCK) {
    const unsigned char* zr = z2q + (long)row * dp;
    float dot = 0.0f;
    for (int j = 0; j < dp; ++j) {
      dot = fmaf(z1_lds[j], vz_fp8_to_f32(zr[j]), dot);
    }
    const float d2 = fmaxf(n1q + n2[row] - 2.0f * scale * dot, 0.0f);
    const float kv = amp2 * matern52_of_d2(d2);
    k_out[(long)q * n + row] = kv;
    mu_acc = fmaf(kv, alpha[row], mu_acc);
    const float* xr = x + (long)row * d;
    float linf = 0.0f;
    for (int j = 0; j < d; ++j) {
      if (!onehot[j]) linf = fmaxf(linf, fabsf(xq_lds[j] - xr[j]));
    }
    min_linf = fminf(min_linf, linf);
  }
  auto fsum = [](float a, float c) { return a + c; };
  auto fmin_ = [](float a, float c) { return fminf(a, c); };
  float mu = block_reduce(mu_acc, red, fsum, 0.0f);
  if (tid == 0) mu_out[q] = mu;
  __syncthreads();
  float dist = block_reduce(min_linf, red, fmin_, INFINITY);
  if (tid == 0) dist_out[q] = dist;
}

extern "C" void launch_ps_kvec_fp8(
    const float* z1f, const float* n1, const float* xq, const float* x,
    const unsigned char* z2q, const float* n2, const float* alpha,
    const unsigned char* onehot, float* k_ws, float* mu_ws,
    float* dist_ws, int b, int n, int d, int dp, float amp2,
    float scale, hipStream_t stream) {
  hipLaunchKernelGGL(ps_kvec_fp8_kernel, dim3(b), dim3(BLOCK), 0,
                     stream, z1f, n1, xq, x, z2q, n2, alpha, onehot,
                     k_ws, mu_ws, dist_ws, b, n, d, dp, amp2, scale);
}

// Chunked quadform + finalize reuse the fp32 kernels below.

// -- Large-N k-vector: row-split variant --------------------------------
//
// ps_kvec_kernel launches one workgroup per candidate (b=25 -> 10% of
// the chip). At N ~ 10^4 that leaves the k/mu/dist pass latency-bound;
// this variant splits the row range over gridDim.y chunks (b x RC
// workgroups) writing per-chunk mu partial sums / dist partial mins,
// reduced by ps_mu_dist_reduce_kernel.

extern "C" __global__ __launch_bounds__(BLOCK) void
ps_kvec_split_kernel(const float* __restrict__ xq,
                     const float* __restrict__ x,
                     const float* __restrict__ inv_ls,
                     const float* __restrict__ alpha,
                     const unsigned char* __restrict__ onehot,
                     float* __restrict__ k_out,      // (B, N)
                     float* __restrict__ mu_part,    // (B, RC)
                     float* __restrict__ dist_part,  // (B, RC)
                     int b, int n, int d, float amp2, int rchunks) {
  __shared__ float red[8];
  __shared__ float xq_lds[512];
  const int q = blockIdx.x;
  const int rc = blockIdx.y;
  if (q >= b) return;
  const int tid = threadIdx.x;
  for (int j = tid; j < d; j += BLOCK) xq_lds[j] = xq[(long)q * d + j];
  __syncthreads();
  const int r0 = (int)((long)rc * n / rchunks);
  const int r1 = (int)((long)(rc + 1) * n / rchunks);
  float mu_acc = 0.0f;
  float min_linf = INFINITY;
  for (int row = r0 + tid; row < r1; row += BLOCK) {
    const float* xr = x + (long)row * d;
    float d2 = 0.0f, linf = 0.0f;
    for (int j = 0; j < d; ++j) {
      const float diff = xq_lds[j] - xr[j];
      const float z = diff * inv_ls[j];
      d2 = fmaf(z, z, d2);
      if (!onehot[j]) linf = fmaxf(linf, fabsf(diff));
    }
    const float kv = amp2 * matern52_of_d2(d2);
    k_out[(long)q * n + row] = kv;
    mu_acc = fmaf(kv, alpha[row], mu_acc);
    min_linf = fminf(min_linf, linf);
  }
  auto fsum = [](float a, float c) { return a + c; };
  auto fmin_ = [](float a, float c) { return fminf(a, c); };
  float mu = block_reduce(mu_acc, red, fsum, 0.0f);
  if (tid == 0) mu_part[(long)q * rchunks + rc] = mu;
  __syncthreads();
  float dist = block_reduce(min_linf, red, fmin_, INFINITY);
  if (tid == 0) dist_part[(long)q * rchunks + rc] = dist;
}

extern "C" __global__ void
ps_mu_dist_reduce_kernel(const float* __restrict__ mu_part,
                         const float* __restrict__ dist_part,
                         float* __restrict__ mu_out,
                         float* __restrict__ dist_out,
                         int b, int rchunks) {
  const int q = blockIdx.x;
  if (q >= b) return;
  const int tid = threadIdx.x;
  __shared__ float red[8];
  float s = 0.0f, m = INFINITY;
  for (int c = tid; c < rchunks; c += blockDim.x) {
    s += mu_part[(long)q * rchunks + c];
    m = fminf(m, dist_part[(long)q * rchunks + c]);
  }
  auto fsum = [](float a, float c) { return a + c; };
  auto fmin_ = [](float a, float c) { return fminf(a, c); };
  float mu = block_reduce(s, red, fsum, 0.0f);
  if (tid == 0) mu_out[q] = mu;
  __syncthreads();
  float dist = block_reduce(m, red, fmin_, INFINITY);
  if (tid == 0) dist_out[q] = dist;
}

extern "C" void launch_ps_kvec_split(
    const float* xq, const float* x, const float* inv_ls,
    const float* alpha, const unsigned char* onehot, float* k_ws,
    float* mu_part, float* dist_part, float* mu_ws, float* dist_ws,
    int b, int n, int d, float amp2, int rchunks, hipStream_t stream) {
  hipLaunchKernelGGL(ps_kvec_split_kernel, dim3(b, rchunks), dim3(BLOCK),
                     0, stream, xq, x, inv_ls, alpha, onehot, k_ws,
                     mu_part, dist_part, b, n, d, amp2, rchunks);
  hipLaunchKernelGGL(ps_mu_dist_reduce_kernel, dim3(b), dim3(256), 0,
                     stream, mu_part, dist_part, mu_ws, dist_ws, b,
                     rchunks);
}

// -- Tile-based small/medium-N quadform ---------------------------------

extern "C" __global__ __launch_bounds__(BLOCK) void
ps_quadform_tile_kernel(const float* __restrict__ k_in,   // (B, N)
                        const float* __restrict__ kinv,   // (N, N)
                        float* __restrict__ var_part,     // (B, T)
                        int b, int n, int tiles_n) {
  __shared__ float k_i[VZ_QF_QMAX * VZ_QF_TILE];
  __shared__ float k_j[VZ_QF_QMAX * VZ_QF_TILE];
  const int total = tiles_n * tiles_n;
  for (int tile = blockIdx.x; tile < total; tile += gridDim.x) {
    const int t = tile;
    vz_quadform_tile(
        kinv, b, n, t, tiles_n, k_i, k_j,
        [&](long idx) { return k_in[idx]; },
        [&](int q, float v) {
          var_part[(long)q * total + t] = v;
        });
  }
}

extern "C" __global__ void
ps_reduce_parts_kernel(const float* __restrict__ part,
                       float* __restrict__ quad, int b, int n_wgs);

extern "C" void launch_ps_quadform_tile(
    const float* k_ws, const float* kinv, float* var_part, float* quad,
    int b, int n, hipStream_t stream) {
  const int tiles_n = (n + VZ_QF_TILE - 1) / VZ_QF_TILE;
  const int total = tiles_n * tiles_n;
  int grid = total < 2048 ? total : 2048;
  hipLaunchKernelGGL(ps_quadform_tile_kernel, dim3(grid), dim3(BLOCK),
                     0, stream, k_ws, kinv, var_part, b, n, tiles_n);
  hipLaunchKernelGGL(ps_reduce_parts_kernel, dim3(b), dim3(256), 0,
                     stream, var_part, quad, b, total);
}

// -- Multi-objective fused helpers --------------------------------------
//
// The MO (config 5) scorer was ~25 eager launches per Eagle iteration
// (per-metric predicts + the HV-scalarization chain) and the op mix is
// not stream-capturable on ROCm 7.2. These two kernels collapse it to
// 3 launches per metric (k-vec + quadform + mean/std finalize) plus
// ONE scalarize+trust-region kernel.

extern "C" __global__ void
ps_finalize_meanstd_kernel(const float* __restrict__ mu_in,
                           const float* __restrict__ var_part,
                           float* __restrict__ mean_out,
                           float* __restrict__ sd_out, int b,
                           float amp2, float mean_c, int nchunk) {
  const int q = blockIdx.x * blockDim.x + threadIdx.x;
  if (q >= b) return;
  float var = 0.0f;
  for (int c = 0; c < nchunk; ++c) var += var_part[q * nchunk + c];
  var = fmaxf(amp2 - var, 1e-12f);
  mean_out[q] = mu_in[q] + mean_c;
  sd_out[q] = sqrtf(var);
}

extern "C" __global__ void
ps_finalize_meanstd_direct_kernel(const float* __restrict__ mu_in,
                                  const float* __restrict__ quad_in,
                                  float* __restrict__ mean_out,
                                  float* __restrict__ sd_out, int b,
                                  float amp2, float mean_c) {
  const int q = blockIdx.x * blockDim.x + threadIdx.x;
  if (q >= b) return;
  mean_out[q] = mu_in[q] + mean_c;
  sd_out[q] = sqrtf(fmaxf(amp2 - quad_in[q], 1e-12f));
}

// score[q] = mean_s min_m (ucb[m,q] - ref[m]) / w[s,m], then the
// trust-region cutoff using the precomputed min-L-inf distance.
// means/sds are (M, B) row-major; weights (S, M).
extern "C" __global__ __launch_bounds__(BLOCK) void
hv_scalarize_tr_kernel(const float* __restrict__ means,
                       const float* __restrict__ sds,
                       const float* __restrict__ weights,
                       const float* __restrict__ ref,   // (M,) or NULL
                       const float* __restrict__ dist,  // (B,) or NULL
                       float* __restrict__ out, int b, int m, int s,
                       float coef, float tr_radius) {
  __shared__ float red[8];
  const int q = blockIdx.x;
  if (q >= b) return;
  const int tid = threadIdx.x;
  float acc = 0.0f;
  for (int si = tid; si < s; si += BLOCK) {
    float v = INFINITY;
    for (int mi = 0; mi < m; ++mi) {
      float y = means[mi * b + q] + coef * sds[mi * b + q];
      if (ref) y -= ref[mi];
      v = fminf(v, y / weights[si * m + mi]);
    }
    acc += v;
  }
  auto fsum = [](float a, float c) { return a + c; };
  float total = block_reduce(acc, red, fsum, 0.0f);
  if (tid == 0) {
    float score = total / (float)s;
    if (dist && tr_radius > 0.0f && tr_radius <= 0.5f &&
        dist[q] > tr_radius) {
      score = -1e4f - dist[q];
    }
    out[q] = score;
  }
}

extern "C" void launch_ps_finalize_meanstd(
    const float* mu_ws, const float* var_ws, float* mean_out,
    float* sd_out, int b, float amp2, float mean_c, int nchunk,
    hipStream_t stream) {
  const int fin_block = 256;
  hipLaunchKernelGGL(ps_finalize_meanstd_kernel,
                     dim3((b + fin_block - 1) / fin_block),
                     dim3(fin_block), 0, stream, mu_ws, var_ws,
                     mean_out, sd_out, b, amp2, mean_c, nchunk);
}

extern "C" void launch_ps_finalize_meanstd_direct(
    const float* mu_ws, const float* quad, float* mean_out,
    float* sd_out, int b, float amp2, float mean_c,
    hipStream_t stream) {
  const int fin_block = 256;
  hipLaunchKernelGGL(ps_finalize_meanstd_direct_kernel,
                     dim3((b + fin_block - 1) / fin_block),
                     dim3(fin_block), 0, stream, mu_ws, quad,
                     mean_out, sd_out, b, amp2, mean_c);
}

extern "C" void launch_hv_scalarize_tr(
    const float* means, const float* sds, const float* weights,
    const float* ref, const float* dist, float* out, int b, int m,
    int s, float coef, float tr_radius, hipStream_t stream) {
  hipLaunchKernelGGL(hv_scalarize_tr_kernel, dim3(b), dim3(BLOCK), 0,
                     stream, means, sds, weights, ref, dist, out, b, m,
                     s, coef, tr_radius);
}

// -- Large-N quadform: hand-written split-K kernel ----------------------
//
// quad[q] = k_q^T Kinv k_q for b <= 32 candidates at N ~ 10^4
// (BASELINE config 4). rocBLAS's skinny SGEMM (25 x N x N) realizes
// ~1.2 TB/s effective here (0.34 ms/iter measured); the HBM floor is
// one full Kinv read = N^2 * 4 B ~ 50 us. This kernel is shaped for
// that floor on the CDNA4 VECTOR ALU (gfx950 has no fp32 MFMA — guide
// "assuming fp32 MFMA exists" pitfall):
//
//  - each THREAD owns one Kinv column j; for each row i the 256
//    threads of a workgroup read kinv[i][j0..j0+255] -> fully
//    coalesced row segments, Kinv read EXACTLY once grid-wide;
//  - the 32 candidate values k[q][i] come from LDS at the same
//    address for every lane -> conflict-free broadcast reads;
//  - acc[32] registers/thread accumulate t_j[q] partials; the final
//    k[q][j] weighting + block reduction writes one partial per
//    (q, workgroup), summed by ps_finalize_kernel-style pass.
//
// Work split: grid = (JCHUNKS, ICHUNKS); JCHUNKS*ICHUNKS workgroups
// fill the 256 CUs (XCD-aware linearization not needed: each block's
// stream is long and bandwidth-shaped).

#define QF_QMAX 32
// 256-float tiles keep k_lds at 32 KB -> 4 workgroups/CU resident
// (64 KB tiles measured 2 waves/SIMD and 440 GB/s; see
// profiles/quadform_ab_r2.json).
#define QF_ITILE 256

extern "C" __global__ __launch_bounds__(BLOCK) void
ps_quadform_big_kernel(const float* __restrict__ k_in,   // (B, N)
                       const float* __restrict__ kinv,   // (N, N)
                       float* __restrict__ part,  // (B, n_wgs)
                       int b, int n, int ichunks) {
  __shared__ float k_lds[QF_QMAX][QF_ITILE];
  __shared__ float red_lds[QF_QMAX][BLOCK / WAVE_SIZE];
  const int tid = threadIdx.x;
  const int jc = blockIdx.x;          // j-chunk of 256 columns
  const int ic = blockIdx.y;          // i-panel
  const int n_wgs = gridDim.x * gridDim.y;
  const int wg = jc * ichunks + ic;
  const int j = jc * BLOCK + tid;     // this thread's column
  const bool j_ok = j < n;
  const long i0 = (long)ic * n / ichunks;
  const long i1 = (long)(ic + 1) * n / ichunks;

  float acc[QF_QMAX];
#pragma unroll
  for (int q = 0; q < QF_QMAX; ++q) acc[q] = 0.0f;

  for (long t0 = i0; t0 < i1; t0 += QF_ITILE) {
    const int tlen = (int)min((long)QF_ITILE, i1 - t0);
    // Cooperative k staging: rows q < b from global, zeros above so
    // the unrolled q-loop below needs no bound check.
    for (int e = tid; e < QF_QMAX * QF_ITILE; e += BLOCK) {
      const int q = e / QF_ITILE;
      const int i = e % QF_ITILE;
      k_lds[q][i] = (q < b && i < tlen)
          ? k_in[(long)q * n + t0 + i] : 0.0f;
    }
    __syncthreads();
    if (j_ok) {
      // 4 rows per step: 4 independent global loads in flight per
      // thread + ds_read_b128 broadcasts of the candidate block.
      int i = 0;
      for (; i + 4 <= tlen; i += 4) {
        const long base = (t0 + i) * (long)n + j;
        const float kv0 = kinv[base];
        const float kv1 = kinv[base + n];
        const float kv2 = kinv[base + 2L * n];
        const float kv3 = kinv[base + 3L * n];
#pragma unroll
        for (int q = 0; q < QF_QMAX; ++q) {
          const float4 kq =
              *reinterpret_cast<const float4*>(&k_lds[q][i]);
          float a = acc[q];
          a = fmaf(kv0, kq.x, a);
          a = fmaf(kv1, kq.y, a);
          a = fmaf(kv2, kq.z, a);
          a = fmaf(kv3, kq.w, a);
          acc[q] = a;
        }
      }
      for (; i < tlen; ++i) {
        const float kv = kinv[(t0 + i) * (long)n + j];
#pragma unroll
        for (int q = 0; q < QF_QMAX; ++q) {
          acc[q] = fmaf(kv, k_lds[q][i], acc[q]);
        }
      }
    }
    __syncthreads();
  }

  // Weight by k[q][j] and reduce across the workgroup per q.
  const int wave = tid / WAVE_SIZE;
  const int lane = tid % WAVE_SIZE;
#pragma unroll
  for (int q = 0; q < QF_QMAX; ++q) {
    float v = 0.0f;
    if (j_ok && q < b) v = acc[q] * k_in[(long)q * n + j];
    v = wave_reduce_sum(v);
    if (lane == 0) red_lds[q][wave] = v;
  }
  __syncthreads();
  if (tid < QF_QMAX) {
    float s = 0.0f;
#pragma unroll
    for (int w = 0; w < BLOCK / WAVE_SIZE; ++w) s += red_lds[tid][w];
    if (tid < b) part[(long)tid * n_wgs + wg] = s;
  }
}

extern "C" __global__ void
ps_reduce_parts_kernel(const float* __restrict__ part,  // (B, n_wgs)
                       float* __restrict__ quad,        // (B,)
                       int b, int n_wgs) {
  const int q = blockIdx.x;
  if (q >= b) return;
  const int tid = threadIdx.x;
  __shared__ float red[8];
  float s = 0.0f;
  for (int w = tid; w < n_wgs; w += blockDim.x) {
    s += part[(long)q * n_wgs + w];
  }
  auto fsum = [](float a, float c) { return a + c; };
  float total = block_reduce(s, red, fsum, 0.0f);
  if (tid == 0) quad[q] = total;
}

extern "C" void launch_ps_quadform_big(
    const float* k_ws, const float* kinv, float* part, float* quad,
    int b, int n, hipStream_t stream) {
  const int jchunks = (n + BLOCK - 1) / BLOCK;
  // Fill the chip: >= 512 workgroups, each with a long i-stream.
  int ichunks = (512 + jchunks - 1) / jchunks;
  if (ichunks < 1) ichunks = 1;
  hipLaunchKernelGGL(ps_quadform_big_kernel, dim3(jchunks, ichunks),
                     dim3(BLOCK), 0, stream, k_ws, kinv, part, b, n,
                     ichunks);
  hipLaunchKernelGGL(ps_reduce_parts_kernel, dim3(b), dim3(256), 0,
                     stream, part, quad, b, jchunks * ichunks);
}

// Finalize variant consuming a precomputed (b,) quadform directly.
//
// At huge N the per-candidate K^-1 streaming in ps_quadform_kernel
// re-reads K_inv once per candidate per iteration (b * N^2 * 4 bytes =
// 10 GB at N=10^4, B=25 — none of it L2/MALL-resident at that size),
// which measured 4.2 ms per Eagle iteration on config 4. The host
// instead computes T = k @ K_inv with ONE rocBLAS SGEMM (K_inv read
// once, candidates reused from registers/LDS by the GEMM tiling) and
// quad[q] = sum_i k[q,i] T[q,i]; this kernel applies the acquisition +
// trust region to that.
extern "C" __global__ void
ps_finalize_direct_kernel(const float* __restrict__ mu_in,
                          const float* __restrict__ dist_in,
                          const float* __restrict__ quad_in,
                          float* __restrict__ out, int b, float amp2,
                          float mean_c, int acq, float coef,
                          float best_value, float tr_radius) {
  const int q = blockIdx.x * blockDim.x + threadIdx.x;
  if (q >= b) return;
  float var = fmaxf(amp2 - quad_in[q], 1e-12f);
  const float sd = sqrtf(var);
  const float mu = mu_in[q] + mean_c;
  float score;
  switch (acq) {
    case ACQ_LCB: score = mu - coef * sd; break;
    case ACQ_EI: {
      const float z = (mu - best_value) / sd;
      score = sd * (z * normal_cdf(z) + normal_pdf(z));
      break;
    }
    case ACQ_PI: {
      const float z = (mu - best_value) / sd;
      score = normal_cdf(z);
      break;
    }
    case ACQ_MEAN: score = mu; break;
    case ACQ_STDDEV: score = sd; break;
    case ACQ_UCB:
    default: score = mu + coef * sd; break;
  }
  const float dist = dist_in[q];
  if (tr_radius > 0.0f && tr_radius <= 0.5f && dist > tr_radius) {
    score = -1e4f - dist;
  }
  out[q] = score;
}

extern "C" void launch_ps_kvec(
    const float* xq, const float* x, const float* inv_ls,
    const float* alpha, const unsigned char* onehot, float* k_ws,
    float* mu_ws, float* dist_ws, int b, int n, int d, float amp2,
    hipStream_t stream) {
  hipLaunchKernelGGL(ps_kvec_kernel, dim3(b), dim3(BLOCK), 0, stream, xq,
                     x, inv_ls, alpha, onehot, k_ws, mu_ws, dist_ws, b, n,
                     d, amp2);
}

extern "C" void launch_ps_finalize_direct(
    const float* mu_ws, const float* dist_ws, const float* quad,
    float* out, int b, float amp2, float mean_c, int acq, float coef,
    float best_value, float tr_radius, hipStream_t stream) {
  const int fin_block = 256;
  hipLaunchKernelGGL(ps_finalize_direct_kernel,
                     dim3((b + fin_block - 1) / fin_block),
                     dim3(fin_block), 0, stream, mu_ws, dist_ws, quad,
                     out, b, amp2, mean_c, acq, coef, best_value,
                     tr_radius);
}

extern "C" void launch_ps_quadform_kernel_only(
    const float* k_ws, const float* kinv, float* var_ws, int b, int n,
    hipStream_t stream) {
  hipLaunchKernelGGL(ps_quadform_kernel, dim3(b, NCHUNK), dim3(BLOCK), 0,
                     stream, k_ws, kinv, var_ws, b, n);
}

extern "C" void launch_ps_quadform_finalize(
    const float* k_ws, const float* kinv, const float* mu_ws,
    const float* dist_ws, float* var_ws, float* out, int b, int n,
    float amp2, float mean_c, int acq, float coef, float best_value,
    float tr_radius, hipStream_t stream) {
  hipLaunchKernelGGL(ps_quadform_kernel, dim3(b, NCHUNK), dim3(BLOCK), 0,
                     stream, k_ws, kinv, var_ws, b, n);
  const int fin_block = 256;
  hipLaunchKernelGGL(ps_finalize_kernel,
                     dim3((b + fin_block - 1) / fin_block),
                     dim3(fin_block), 0, stream, mu_ws, dist_ws, var_ws,
                     out, b, amp2, mean_c, acq, coef, best_value,
                     tr_radius);
}

extern "C" void launch_posterior_score_chunked(
    const float* xq, const float* x, const float* inv_ls,
    const float* alpha, const float* kinv, const unsigned char* onehot,
    float* k_ws, float* mu_ws, float* dist_ws, float* var_ws, float* out,
    int b, int n, int d, float amp2, float mean_c, int acq, float coef,
    float best_value, float tr_radius, hipStream_t stream) {
  hipLaunchKernelGGL(ps_kvec_kernel, dim3(b), dim3(BLOCK), 0, stream, xq,
                     x, inv_ls, alpha, onehot, k_ws, mu_ws, dist_ws, b, n,
                     d, amp2);
  hipLaunchKernelGGL(ps_quadform_kernel, dim3(b, NCHUNK), dim3(BLOCK), 0,
                     stream, k_ws, kinv, var_ws, b, n);
  const int fin_block = 256;
  hipLaunchKernelGGL(ps_finalize_kernel,
                     dim3((b + fin_block - 1) / fin_block),
                     dim3(fin_block), 0, stream, mu_ws, dist_ws, var_ws,
                     out, b, amp2, mean_c, acq, coef, best_value,
                     tr_radius);
}
