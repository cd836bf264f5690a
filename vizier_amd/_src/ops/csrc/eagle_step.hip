// Fused Eagle-strategy suggest/update kernels for gfx950.
//
// One workgroup per batch firefly. Replaces ~30 eager tensor launches
// per Eagle iteration with two: suggest (distances -> forces -> moves ->
// perturbation + categorical sampling) and update (accept/penalize/trim).
// The pool (<= 100 flies x D features) is small enough that all inputs
// sit in L2; the kernels are launch-latency-bound by design, which is
// exactly what fusing minimizes.
//
// Algorithm spec: vizier_amd/_src/algorithms/optimizers/eagle.py (the
// torch implementation is the CPU oracle; RNG streams differ).

#include <hip/hip_runtime.h>
#include "common.h"

#define BLOCK 256
#define MAX_POOL 128
#define MAX_Q 16

// -- counter-based RNG (splitmix64 -> uniform in (0,1)) ---------------------

#define splitmix64 vz_splitmix64
#define rng_uniform vz_rng_uniform
#define rng_laplace vz_rng_laplace

// -- suggest -----------------------------------------------------------------

extern "C" __global__ __launch_bounds__(BLOCK) void
eagle_suggest_kernel(const float* __restrict__ pool_cont,  // (P, q, Dc)
                     const long* __restrict__ pool_cat,    // (P, q, Dcat)
                     const float* __restrict__ rewards,    // (P,)
                     const float* __restrict__ perturbations,  // (P,)
                     const long* __restrict__ cat_sizes,   // (Dcat,)
                     float* __restrict__ out_cont,         // (B, q, Dc)
                     long* __restrict__ out_cat,           // (B, q, Dcat)
                     const unsigned long long* __restrict__ iter_ptr,
                     int n_batches, int batch_size, int pool_size, int q,
                     int dc, int dcat, int max_cat, float visibility,
                     float gravity, float neg_gravity, float norm_scale,
                     float cat_factor, float p_same,
                     unsigned long long seed) {
  // Counter handshake (hipGraph-capturable, race-free): suggest reads
  // slot 0 and publishes the iteration to slot 1 for the update kernel;
  // update reads slot 1 and advances slot 0 for the next suggest. All
  // cross-kernel ordering comes from stream order; within a kernel only
  // one block writes, and readers never read the slot their own grid
  // writes.
  const unsigned long long offset = iter_ptr[0];
  const int batch_start = (int)(offset % (unsigned long long)n_batches) *
                          batch_size;
  if (blockIdx.x == 0 && threadIdx.x == 0) {
    ((unsigned long long*)iter_ptr)[1] = offset;
  }
  __shared__ float scale[MAX_POOL];
  __shared__ float red[8];
  __shared__ float scale_sum_s;
  __shared__ float noise_norm[MAX_Q];  // scratch per-dim normalization

  const int b = blockIdx.x;
  if (b >= batch_size) return;
  const int me = batch_start + b;      // pool index of this firefly
  const int tid = threadIdx.x;
  const int n_features = dc + dcat;
  const int flat_c = q * dc;
  const float my_reward = rewards[me];
  const float my_pert = perturbations[me];

  // Phase 1: per-pool-member force -> scale[p].
  for (int p = tid; p < pool_size; p += BLOCK) {
    float d2 = 0.0f;
    for (int j = 0; j < flat_c; ++j) {
      const float diff = pool_cont[me * flat_c + j] -
                         pool_cont[p * flat_c + j];
      d2 = fmaf(diff, diff, d2);
    }
    for (int j = 0; j < q * dcat; ++j) {
      d2 += (pool_cat[me * q * dcat + j] != pool_cat[p * q * dcat + j])
                ? 1.0f : 0.0f;
    }
    const float reward_p = rewards[p];
    const float dir = (reward_p - my_reward >= 0.0f) ? gravity
                                                     : -neg_gravity;
    const float force = __expf(-visibility * d2 / n_features * 10.0f);
    float s = dir * force;
    if (!isfinite(reward_p)) s = 0.0f;
    scale[p] = s;
  }
  __syncthreads();

  // Counts for MEAN normalization.
  float pulls = 0.0f, pushes = 0.0f;
  for (int p = tid; p < pool_size; p += BLOCK) {
    pulls += (scale[p] > 0.0f) ? 1.0f : 0.0f;
    pushes += (scale[p] < 0.0f) ? 1.0f : 0.0f;
  }
  auto fsum = [](float a, float c) { return a + c; };
  float n_pull = block_reduce(pulls, red, fsum, 0.0f);
  if (tid == 0) red[4] = fmaxf(n_pull, 1.0f);
  __syncthreads();
  float n_push = block_reduce(pushes, red, fsum, 0.0f);
  if (tid == 0) red[5] = fmaxf(n_push, 1.0f);
  __syncthreads();
  const float inv_pull = 1.0f / red[4];
  const float inv_push = 1.0f / red[5];

  // Normalize scale in place; accumulate its sum.
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
  if (tid == 0) scale_sum_s = scale_sum;
  __syncthreads();
  scale_sum = scale_sum_s;

  // Phase 2: continuous move + laplace perturbation. Threads over dims.
  for (int j = tid; j < flat_c; j += BLOCK) {
    float moved = 0.0f;
    for (int p = 0; p < pool_size; ++p) {
      moved = fmaf(scale[p], pool_cont[p * flat_c + j], moved);
    }
    const float mine = pool_cont[me * flat_c + j];
    moved = mine + (moved - mine * scale_sum);
    // Laplace noise normalized over the q axis for this (b, dim).
    const int dim = j % dc;
    float noise = rng_laplace(seed, offset,
                              (unsigned)(b * flat_c + j));
    if (q > 1) {
      float maxabs = 1e-12f;
      for (int qq = 0; qq < q; ++qq) {
        const float nv = rng_laplace(seed, offset,
                                     (unsigned)(b * flat_c + qq * dc + dim));
        maxabs = fmaxf(maxabs, fabsf(nv));
      }
      noise /= maxabs;
    } else {
      noise = (noise >= 0.0f) ? 1.0f : -1.0f;  // |laplace|/max == 1
    }
    out_cont[b * flat_c + j] = moved + noise * my_pert;
  }

  // Phase 3: categorical mutation. Threads over (q, dcat) pairs.
  for (int jd = tid; jd < q * dcat; jd += BLOCK) {
    const int dimc = jd % dcat;
    const long size = cat_sizes[dimc];
    const float logit_same = __logf(p_same);
    const float logit_diff = __logf((1.0f - p_same) /
                                    fmaxf((float)size - 1.0f, 1e-9f));
    const long cur = pool_cat[me * q * dcat + jd];
    const float pert_noise =
        rng_laplace(seed, offset ^ 0x9e01ull,
                    (unsigned)(b * q * dcat + jd)) *
        cat_factor * my_pert;
    float best_val = -INFINITY;
    long best_cat = cur;
    for (long c = 0; c < size; ++c) {
      float ssum_c = 0.0f;
      for (int p = 0; p < pool_size; ++p) {
        if (pool_cat[p * q * dcat + jd] == c) ssum_c += scale[p];
      }
      // logits[c] = sum_p scale[p][cat_p == c] + logit_diff, and at the
      // current category: + (-sum(scale) + logit_same - logit_diff).
      float logit = ssum_c + logit_diff;
      if (c == cur) logit += -scale_sum_s + logit_same - logit_diff;
      logit += pert_noise;
      // Gumbel-max sampling.
      const float u = rng_uniform(seed, offset ^ 0x77aaull,
                                  (unsigned)((b * q * dcat + jd) *
                                             max_cat + (int)c));
      const float g = -__logf(-__logf(fmaxf(u, 1e-20f)));
      if (logit + g > best_val) {
        best_val = logit + g;
        best_cat = c;
      }
    }
    out_cat[b * q * dcat + jd] = best_cat;
  }
}

// -- update ------------------------------------------------------------------

extern "C" __global__ __launch_bounds__(BLOCK) void
eagle_update_kernel(float* __restrict__ pool_cont,
                    long* __restrict__ pool_cat,
                    float* __restrict__ rewards,
                    float* __restrict__ perturbations,
                    const float* __restrict__ batch_cont,
                    const long* __restrict__ batch_cat,
                    const float* __restrict__ batch_rewards,
                    const long* __restrict__ cat_sizes,
                    float* __restrict__ best_reward,
                    unsigned long long* __restrict__ iter_ptr,
                    int n_batches, int batch_size, int q, int dc, int dcat,
                    float penalize_factor, float perturbation_lower_bound,
                    float base_perturbation, unsigned long long seed) {
  // One block per batch member. Every block recomputes the (tiny)
  // batch max + old best, so new_best is identical everywhere and
  // writes are idempotent — no cross-block ordering needed.
  const unsigned long long offset = iter_ptr[1];
  const int batch_start = (int)(offset % (unsigned long long)n_batches) *
                          batch_size;
  __shared__ float red[8];
  __shared__ float new_best_s;
  const int tid = threadIdx.x;
  const int i = blockIdx.x;
  if (i >= batch_size) return;

  float local_max = -INFINITY;
  for (int t = tid; t < batch_size; t += BLOCK) {
    local_max = fmaxf(local_max, batch_rewards[t]);
  }
  auto fmax_ = [](float a, float c) { return fmaxf(a, c); };
  float batch_max = block_reduce(local_max, red, fmax_, -INFINITY);
  if (tid == 0) {
    new_best_s = fmaxf(best_reward[0], batch_max);
    if (i == 0) best_reward[0] = new_best_s;
  }
  __syncthreads();
  const float new_best = new_best_s;
  const int flat_c = q * dc;
  const int flat_k = q * dcat;

  const int me = batch_start + i;
  const float new_r = batch_rewards[i];
  const float old_r = rewards[me];
  const bool improved = new_r > old_r;
  float pert = improved ? perturbations[me]
                        : perturbations[me] * penalize_factor;
  float reward = improved ? new_r : old_r;
  const bool dead = (pert < perturbation_lower_bound) &&
                    (reward != new_best);
  if (improved) {
    for (int j = tid; j < flat_c; j += BLOCK) {
      pool_cont[me * flat_c + j] = batch_cont[i * flat_c + j];
    }
    for (int j = tid; j < flat_k; j += BLOCK) {
      pool_cat[me * flat_k + j] = batch_cat[i * flat_k + j];
    }
  }
  if (dead) {
    for (int j = tid; j < flat_c; j += BLOCK) {
      pool_cont[me * flat_c + j] = rng_uniform(
          seed, offset ^ 0x5151ull, (unsigned)(me * flat_c + j));
    }
    for (int j = tid; j < flat_k; j += BLOCK) {
      const long size = cat_sizes[j % dcat];
      const float u = rng_uniform(seed, offset ^ 0x1234ull,
                                  (unsigned)(me * flat_k + j));
      long c = (long)(u * size);
      pool_cat[me * flat_k + j] = min(c, size - 1);
    }
    reward = -INFINITY;
    pert = base_perturbation;
  }
  if (tid == 0) {
    rewards[me] = reward;
    perturbations[me] = pert;
    if (i == 0) iter_ptr[0] = offset + 1;  // next suggest's iteration
  }
}

// -- launchers ---------------------------------------------------------------

extern "C" void launch_eagle_suggest(
    const float* pool_cont, const long* pool_cat, const float* rewards,
    const float* perturbations, const long* cat_sizes, float* out_cont,
    long* out_cat, const unsigned long long* iter_ptr, int n_batches,
    int batch_size, int pool_size, int q,
    int dc, int dcat, int max_cat, float visibility, float gravity,
    float neg_gravity, float norm_scale, float cat_factor, float p_same,
    unsigned long long seed, hipStream_t stream) {
  hipLaunchKernelGGL(eagle_suggest_kernel, dim3(batch_size), dim3(BLOCK),
                     0, stream, pool_cont, pool_cat, rewards,
                     perturbations, cat_sizes, out_cont, out_cat,
                     iter_ptr, n_batches, batch_size, pool_size, q, dc,
                     dcat, max_cat, visibility, gravity, neg_gravity,
                     norm_scale, cat_factor, p_same, seed);
}

extern "C" void launch_eagle_update(
    float* pool_cont, long* pool_cat, float* rewards, float* perturbations,
    const float* batch_cont, const long* batch_cat,
    const float* batch_rewards, const long* cat_sizes, float* best_reward,
    unsigned long long* iter_ptr, int n_batches, int batch_size, int q,
    int dc, int dcat,
    float penalize_factor, float perturbation_lower_bound,
    float base_perturbation, unsigned long long seed, hipStream_t stream) {
  hipLaunchKernelGGL(eagle_update_kernel, dim3(batch_size), dim3(BLOCK),
                     0, stream,
                     pool_cont, pool_cat, rewards, perturbations,
                     batch_cont, batch_cat, batch_rewards, cat_sizes,
                     best_reward, iter_ptr, n_batches, batch_size, q, dc,
                     dcat, penalize_factor, perturbation_lower_bound,
                     base_perturbation, seed);
}
