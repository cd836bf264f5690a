// Persistent cooperative Eagle-sweep megakernel for gfx950.
//
// Runs the ENTIRE steady-state acquisition sweep (suggest -> GP
// posterior score -> update, x thousands of iterations) inside ONE
// cooperatively-launched kernel with grid-wide barriers between
// phases. The hipGraph path replays ~5 kernels per iteration at
// ~85 us/iteration — dominated by kernel dispatch, not work (the
// quadform reads only ~4 MB of K^-1 per iteration, microseconds at
// HBM3E speed). Keeping the loop device-side removes every dispatch.
//
// Scope: continuous-only spaces, q == 1 (the flagship GP-Bandit
// config); the optimizer falls back to the hipGraph path otherwise.
//
// Determinism: phases replicate the standalone kernels VERBATIM —
// same counter-based RNG streams (common.h), same NCHUNK quadform
// partition, same block-reduce orders — so a sweep through this
// kernel is BIT-IDENTICAL to the hipGraph path with the same seeds
// (tested in tests/test_gpu_ops.py).
//
// Deadlock safety: every phase is a grid-stride loop, so any grid
// size >= 1 is correct; the host launcher clamps the grid to the
// cooperative-launch occupancy limit.

#include <hip/hip_cooperative_groups.h>
#include <hip/hip_runtime.h>

#include "common.h"

#define BLOCK 256
#define MAX_POOL 128
#define SWEEP_NCHUNK 10  // must match posterior_score.hip's NCHUNK

// acquisition codes (must match posterior_score.hip)
#define ACQ_UCB 0
#define ACQ_LCB 1
#define ACQ_EI 2
#define ACQ_PI 3
#define ACQ_MEAN 4
#define ACQ_STDDEV 5

namespace cg = cooperative_groups;

extern "C" __global__ __launch_bounds__(BLOCK) void
eagle_sweep_kernel(
    // Eagle pool state (updated in place)
    float* __restrict__ pool_cont,         // (P, Dc)
    float* __restrict__ rewards,           // (P,)
    float* __restrict__ perturbations,     // (P,)
    float* __restrict__ best_reward,       // (1,)
    unsigned long long* __restrict__ iter_ptr,  // (2,) kept consistent
    // GP scorer state (read-only)
    const float* __restrict__ x,           // (N, D) training features
    const float* __restrict__ inv_ls,      // (D,)
    const float* __restrict__ alpha,       // (N,)
    const float* __restrict__ kinv,        // (N, N)
    // workspaces
    float* __restrict__ out_cont,          // (B, Dc) candidates
    float* __restrict__ k_ws,              // (B, N)
    float* __restrict__ mu_ws,             // (B,)
    float* __restrict__ dist_ws,           // (B,)
    float* __restrict__ var_ws,            // (B, SWEEP_NCHUNK)
    float* __restrict__ scores,            // (B,)
    // shapes / iteration range
    int n_batches, int batch_size, int pool_size, int dc, int n,
    long long it_start, long long iterations,
    // eagle config
    float visibility, float gravity, float neg_gravity, float norm_scale,
    float penalize_factor, float perturbation_lower_bound,
    float base_perturbation,
    unsigned long long seed_suggest, unsigned long long seed_update,
    // scorer config
    float amp2, float mean_c, int acq, float coef, float best_value,
    float tr_radius) {
  cg::grid_group grid = cg::this_grid();
  __shared__ float scale[MAX_POOL];
  __shared__ float red[8];
  __shared__ float s_scalar;
  __shared__ float xq_lds[512];

  const int tid = threadIdx.x;
  const int G = gridDim.x;
  auto fsum = [](float a, float c) { return a + c; };
  auto fmin_ = [](float a, float c) { return fminf(a, c); };
  auto fmax_ = [](float a, float c) { return fmaxf(a, c); };

  for (long long it = it_start; it < it_start + iterations; ++it) {
    const unsigned long long offset = (unsigned long long)it;
    const int batch_start =
        (int)(offset % (unsigned long long)n_batches) * batch_size;

    // ---- Phase A: suggest (one workgroup per candidate) ----
    for (int b = blockIdx.x; b < batch_size; b += G) {
      const int me = batch_start + b;
      const float my_reward = rewards[me];
      const float my_pert = perturbations[me];
      for (int p = tid; p < pool_size; p += BLOCK) {
        float d2 = 0.0f;
        for (int j = 0; j < dc; ++j) {
          const float diff = pool_cont[me * dc + j] - pool_cont[p * dc + j];
          d2 = fmaf(diff, diff, d2);
        }
        const float reward_p = rewards[p];
        const float dir = (reward_p - my_reward >= 0.0f) ? gravity
                                                         : -neg_gravity;
        const float force = __expf(-visibility * d2 / dc * 10.0f);
        float s = dir * force;
        if (!isfinite(reward_p)) s = 0.0f;
        scale[p] = s;
      }
      __syncthreads();
      float pulls = 0.0f, pushes = 0.0f;
      for (int p = tid; p < pool_size; p += BLOCK) {
        pulls += (scale[p] > 0.0f) ? 1.0f : 0.0f;
        pushes += (scale[p] < 0.0f) ? 1.0f : 0.0f;
      }
      float n_pull = block_reduce(pulls, red, fsum, 0.0f);
      if (tid == 0) red[4] = fmaxf(n_pull, 1.0f);
      __syncthreads();
      float n_push = block_reduce(pushes, red, fsum, 0.0f);
      if (tid == 0) red[5] = fmaxf(n_push, 1.0f);
      __syncthreads();
      const float inv_pull = 1.0f / red[4];
      const float inv_push = 1.0f / red[5];
      float ssum = 0.0f;
      for (int p = tid; p < pool_size; p += BLOCK) {
        const float s = scale[p];
        const float ns = norm_scale * (s > 0.0f ? s * inv_pull
                                                : s * inv_push);
        scale[p] = ns;
        ssum += ns;
      }
      __syncthreads();
      float scale_sum = block_reduce(ssum, red, fsum, 0.0f);
      if (tid == 0) s_scalar = scale_sum;
      __syncthreads();
      scale_sum = s_scalar;
      for (int j = tid; j < dc; j += BLOCK) {
        float moved = 0.0f;
        for (int p = 0; p < pool_size; ++p) {
          moved = fmaf(scale[p], pool_cont[p * dc + j], moved);
        }
        const float mine = pool_cont[me * dc + j];
        moved = mine + (moved - mine * scale_sum);
        float noise = vz_rng_laplace(seed_suggest, offset,
                                     (unsigned)(b * dc + j));
        noise = (noise >= 0.0f) ? 1.0f : -1.0f;  // q == 1
        out_cont[b * dc + j] = moved + noise * my_pert;
      }
      __syncthreads();
    }
    __threadfence();
    grid.sync();

    // ---- Phase B1: k-vectors + mu + trust-region distance ----
    for (int q = blockIdx.x; q < batch_size; q += G) {
      for (int j = tid; j < dc; j += BLOCK) xq_lds[j] = out_cont[q * dc + j];
      __syncthreads();
      float mu_acc = 0.0f;
      float min_linf = INFINITY;
      for (int row = tid; row < n; row += BLOCK) {
        const float* xr = x + (long)row * dc;
        float d2 = 0.0f, linf = 0.0f;
        for (int j = 0; j < dc; ++j) {
          const float diff = xq_lds[j] - xr[j];
          const float z = diff * inv_ls[j];
          d2 = fmaf(z, z, d2);
          linf = fmaxf(linf, fabsf(diff));  // continuous-only: no onehot
        }
        const float kv = amp2 * matern52_of_d2(d2);
        k_ws[(long)q * n + row] = kv;
        mu_acc = fmaf(kv, alpha[row], mu_acc);
        min_linf = fminf(min_linf, linf);
      }
      float mu = block_reduce(mu_acc, red, fsum, 0.0f);
      if (tid == 0) mu_ws[q] = mu;
      __syncthreads();
      float dist = block_reduce(min_linf, red, fmin_, INFINITY);
      if (tid == 0) dist_ws[q] = dist;
      __syncthreads();
    }
    __threadfence();
    grid.sync();

    // ---- Phase B2: K^-1 quadform (wave-per-row, NCHUNK partition) ----
    const int wave = tid / WAVE_SIZE;
    const int lane = tid % WAVE_SIZE;
    const int waves = BLOCK / WAVE_SIZE;
    const int n4 = n / 4;
    for (int pair = blockIdx.x; pair < batch_size * SWEEP_NCHUNK;
         pair += G) {
      const int q = pair % batch_size;
      const int chunk = pair / batch_size;
      const int j0 = (int)((long)chunk * n / SWEEP_NCHUNK);
      const int j1 = (int)((long)(chunk + 1) * n / SWEEP_NCHUNK);
      const float* k = k_ws + (long)q * n;
      float acc = 0.0f;
      for (int j = j0 + wave; j < j1; j += waves) {
        const float4* row4 =
            reinterpret_cast<const float4*>(kinv + (long)j * n);
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
        for (int i = 4 * n4 + lane; i < n; i += WAVE_SIZE) {
          t_j = fmaf(kinv[(long)j * n + i], k[i], t_j);
        }
        t_j = wave_reduce_sum(t_j);
        if (lane == 0) acc = fmaf(k[j], t_j, acc);
      }
      float v = block_reduce(acc, red, fsum, 0.0f);
      if (tid == 0) var_ws[q * SWEEP_NCHUNK + chunk] = v;
      __syncthreads();
    }
    __threadfence();
    grid.sync();

    // ---- Phase B3: finalize scores (one workgroup) ----
    if (blockIdx.x == 0) {
      for (int q = tid; q < batch_size; q += BLOCK) {
        float var = 0.0f;
        for (int c = 0; c < SWEEP_NCHUNK; ++c)
          var += var_ws[q * SWEEP_NCHUNK + c];
        var = fmaxf(amp2 - var, 1e-12f);
        const float sd = sqrtf(var);
        const float mu = mu_ws[q] + mean_c;
        float score;
        switch (acq) {
          case ACQ_LCB: score = mu - coef * sd; break;
          case ACQ_EI: {
            const float z = (mu - best_value) / sd;
            score = sd * (z * vz_normal_cdf(z) + vz_normal_pdf(z));
            break;
          }
          case ACQ_PI: {
            const float z = (mu - best_value) / sd;
            score = vz_normal_cdf(z);
            break;
          }
          case ACQ_MEAN: score = mu; break;
          case ACQ_STDDEV: score = sd; break;
          case ACQ_UCB:
          default: score = mu + coef * sd; break;
        }
        const float dist = dist_ws[q];
        if (tr_radius > 0.0f && tr_radius <= 0.5f && dist > tr_radius) {
          score = -1e4f - dist;
        }
        scores[q] = score;
      }
    }
    __threadfence();
    grid.sync();

    // ---- Phase C: update (one workgroup per batch member) ----
    for (int i = blockIdx.x; i < batch_size; i += G) {
      float local_max = -INFINITY;
      for (int t = tid; t < batch_size; t += BLOCK) {
        local_max = fmaxf(local_max, scores[t]);
      }
      float batch_max = block_reduce(local_max, red, fmax_, -INFINITY);
      if (tid == 0) {
        s_scalar = fmaxf(best_reward[0], batch_max);
        if (i == 0) best_reward[0] = s_scalar;
      }
      __syncthreads();
      const float new_best = s_scalar;
      const int me = batch_start + i;
      const float new_r = scores[i];
      const float old_r = rewards[me];
      const bool improved = new_r > old_r;
      float pert = improved ? perturbations[me]
                            : perturbations[me] * penalize_factor;
      float reward = improved ? new_r : old_r;
      const bool dead = (pert < perturbation_lower_bound) &&
                        (reward != new_best);
      if (improved) {
        for (int j = tid; j < dc; j += BLOCK) {
          pool_cont[me * dc + j] = out_cont[i * dc + j];
        }
      }
      if (dead) {
        for (int j = tid; j < dc; j += BLOCK) {
          pool_cont[me * dc + j] = vz_rng_uniform(
              seed_update, offset ^ 0x5151ull, (unsigned)(me * dc + j));
        }
        reward = -INFINITY;
        pert = base_perturbation;
      }
      if (tid == 0) {
        rewards[me] = reward;
        perturbations[me] = pert;
      }
      __syncthreads();
    }
    __threadfence();
    grid.sync();
  }

  if (blockIdx.x == 0 && tid == 0) {
    iter_ptr[0] = (unsigned long long)(it_start + iterations);
    iter_ptr[1] = (unsigned long long)(it_start + iterations - 1);
  }
}

// Host launcher: cooperative launch with occupancy-clamped grid.
// Returns the grid size used, 0 if cooperative launch is unavailable,
// or a negative hipError_t on failure.
extern "C" int launch_eagle_sweep(
    float* pool_cont, float* rewards, float* perturbations,
    float* best_reward, unsigned long long* iter_ptr, const float* x,
    const float* inv_ls, const float* alpha, const float* kinv,
    float* out_cont, float* k_ws, float* mu_ws, float* dist_ws,
    float* var_ws, float* scores, int n_batches, int batch_size,
    int pool_size, int dc, int n, long long it_start, long long iterations,
    float visibility, float gravity, float neg_gravity, float norm_scale,
    float penalize_factor, float perturbation_lower_bound,
    float base_perturbation, unsigned long long seed_suggest,
    unsigned long long seed_update, float amp2, float mean_c, int acq,
    float coef, float best_value, float tr_radius, hipStream_t stream) {
  int dev = 0;
  (void)hipGetDevice(&dev);
  int coop = 0;
  (void)hipDeviceGetAttribute(&coop, hipDeviceAttributeCooperativeLaunch,
                              dev);
  if (!coop) return 0;
  int num_cu = 0;
  (void)hipDeviceGetAttribute(&num_cu,
                              hipDeviceAttributeMultiprocessorCount, dev);
  int blocks_per_cu = 0;
  hipError_t err = hipOccupancyMaxActiveBlocksPerMultiprocessor(
      &blocks_per_cu, (const void*)eagle_sweep_kernel, BLOCK, 0);
  if (err != hipSuccess || blocks_per_cu < 1) return 0;
  int grid = batch_size * SWEEP_NCHUNK;           // fills phase B2
  const int max_grid = blocks_per_cu * num_cu;
  if (grid > max_grid) grid = max_grid;
  if (grid < 1) grid = 1;

  void* args[] = {
      &pool_cont, &rewards, &perturbations, &best_reward, &iter_ptr,
      &x, &inv_ls, &alpha, &kinv, &out_cont, &k_ws, &mu_ws, &dist_ws,
      &var_ws, &scores, &n_batches, &batch_size, &pool_size, &dc, &n,
      &it_start, &iterations, &visibility, &gravity, &neg_gravity,
      &norm_scale, &penalize_factor, &perturbation_lower_bound,
      &base_perturbation, &seed_suggest, &seed_update, &amp2, &mean_c,
      &acq, &coef, &best_value, &tr_radius};
  err = hipLaunchCooperativeKernel((const void*)eagle_sweep_kernel,
                                   dim3(grid), dim3(BLOCK), args, 0,
                                   stream);
  if (err != hipSuccess) return -(int)err;
  return grid;
}
