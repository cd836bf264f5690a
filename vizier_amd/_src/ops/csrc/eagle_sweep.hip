// Persistent cooperative Eagle-sweep megakernel for gfx950 (v2).
//
// Runs the ENTIRE steady-state acquisition sweep (suggest -> GP
// posterior score -> update, x thousands of iterations) inside ONE
// cooperatively-launched kernel. The hipGraph path replays ~5 kernels
// per iteration at ~85 us/iteration — dispatch-bound, not work-bound.
//
// MI355X coherence design (the crux): the 8 per-XCD L2s are NOT
// cross-coherent, and generic device-scope fences (cg::grid.sync /
// __threadfence) writeback+invalidate L2 — measured ~65 us per sync,
// which made a v1 of this kernel SLOWER than the graph path. v2
// instead:
//   - routes ONLY the small inter-phase buffers (pool state, k-vector,
//     quadform partials, mu/dist) through agent-scope RELAXED atomic
//     loads/stores — coherent at the memory side, no cache flush;
//   - keeps the heavy read-only data (K^-1, x, alpha) on the normal
//     cached path (L2-resident across all iterations);
//   - uses a hand-rolled sense-reversing grid barrier on agent-scope
//     atomics (s_waitcnt vmcnt(0) before arrival orders the relaxed
//     data stores), 3 barriers per iteration;
//   - merges phases so a single workgroup owns candidate b through
//     suggest+k-vec (A) and finalize+update (C); only the quadform (B)
//     fans out across the grid.
//
// Scope: continuous-only spaces, q == 1 (the flagship GP-Bandit
// config); the optimizer falls back to the hipGraph path otherwise.
//
// Determinism: phases replicate the standalone kernels VERBATIM (same
// counter-based RNG streams, same NCHUNK quadform partition, same
// block-reduce orders), so the sweep is BIT-IDENTICAL to the hipGraph
// path for the same seeds (asserted by tests/test_gpu_ops.py).
//
// Deadlock safety: every phase is a grid-stride loop (any grid size is
// correct); the launcher clamps the grid to the cooperative occupancy
// limit, and cooperative launch guarantees co-residency.

#include <hip/hip_runtime.h>

#include "common.h"

#define BLOCK 256
#define MAX_POOL 128
#define SWEEP_NCHUNK 10   // must match posterior_score.hip's NCHUNK
#define POOL_LDS_CAP 8192  // floats: stage pool in LDS when it fits

#define ACQ_UCB 0
#define ACQ_LCB 1
#define ACQ_EI 2
#define ACQ_PI 3
#define ACQ_MEAN 4
#define ACQ_STDDEV 5

// Agent-scope (device) coherent access helpers: compile to memory-side
// accesses that bypass the incoherent per-XCD L2 path for plain loads.
__device__ __forceinline__ float cload(const float* p) {
  return __hip_atomic_load(p, __ATOMIC_RELAXED, __HIP_MEMORY_SCOPE_AGENT);
}
__device__ __forceinline__ void cstore(float* p, float v) {
  __hip_atomic_store(p, v, __ATOMIC_RELAXED, __HIP_MEMORY_SCOPE_AGENT);
}

// Sense-reversing grid barrier on agent-scope atomics. bar[0] = arrival
// count, bar[1] = generation.
//
// Deliberately RELAXED atomics: acq_rel orderings compile to
// buffer_wbl2 / buffer_inv (full per-XCD L2 writeback+invalidate),
// which costs tens of microseconds per barrier AND evicts the
// L2-resident K^-1 that phase B depends on. Ordering is instead
// established by hand:
//   - every thread drains its own outstanding global ops
//     (s_waitcnt 0) before the block barrier, so all sc1 data stores
//     have reached the coherence point before tid 0 arrives;
//   - all cross-workgroup data moves through sc1 accesses (cload /
//     cstore), which always read/write the coherence point, so no
//     cache invalidation is needed on the acquire side;
//   - waves issue memory ops in order, so the spin-exit load ordering
//     is sufficient (compiler reordering is fenced with asm).
__device__ __forceinline__ void grid_sync(unsigned int* bar) {
  __builtin_amdgcn_s_waitcnt(0);
  __syncthreads();
  if (threadIdx.x == 0) {
    const unsigned int gen =
        __hip_atomic_load(bar + 1, __ATOMIC_RELAXED,
                          __HIP_MEMORY_SCOPE_AGENT);
    asm volatile("" ::: "memory");
    const unsigned int arrived = __hip_atomic_fetch_add(
        bar, 1u, __ATOMIC_RELAXED, __HIP_MEMORY_SCOPE_AGENT);
    if (arrived == gridDim.x - 1) {
      __hip_atomic_store(bar, 0u, __ATOMIC_RELAXED,
                         __HIP_MEMORY_SCOPE_AGENT);
      __builtin_amdgcn_s_waitcnt(0);  // count reset lands before gen++
      __hip_atomic_fetch_add(bar + 1, 1u, __ATOMIC_RELAXED,
                             __HIP_MEMORY_SCOPE_AGENT);
    } else {
      while (__hip_atomic_load(bar + 1, __ATOMIC_RELAXED,
                               __HIP_MEMORY_SCOPE_AGENT) == gen) {
        __builtin_amdgcn_s_sleep(8);
      }
    }
    asm volatile("" ::: "memory");
  }
  __syncthreads();
}

__device__ __forceinline__ float acq_score(int acq, float mu, float sd,
                                           float coef, float best_value) {
  switch (acq) {
    case ACQ_LCB: return mu - coef * sd;
    case ACQ_EI: {
      const float z = (mu - best_value) / sd;
      return sd * (z * vz_normal_cdf(z) + vz_normal_pdf(z));
    }
    case ACQ_PI: return vz_normal_cdf((mu - best_value) / sd);
    case ACQ_MEAN: return mu;
    case ACQ_STDDEV: return sd;
    case ACQ_UCB:
    default: return mu + coef * sd;
  }
}

extern "C" __global__ __launch_bounds__(BLOCK) void
eagle_sweep_kernel(
    float* __restrict__ pool_cont,         // (P, Dc)   [coherent]
    float* __restrict__ rewards,           // (P,)      [coherent]
    float* __restrict__ perturbations,     // (P,)      [coherent]
    float* __restrict__ best_reward,       // (1,)      [coherent]
    unsigned long long* __restrict__ iter_ptr,  // (2,)
    unsigned int* __restrict__ barrier_buf,     // (2,) zeroed
    const float* __restrict__ x,           // (N, D)    [cached]
    const float* __restrict__ inv_ls,      // (D,)      [cached]
    const float* __restrict__ alpha,       // (N,)      [cached]
    const float* __restrict__ kinv,        // (N, N)    [cached]
    float* __restrict__ out_cont,          // (B, Dc)   [wg-private]
    float* __restrict__ k_ws,              // (B, N)    [coherent]
    float* __restrict__ mu_ws,             // (B,)      [coherent]
    float* __restrict__ dist_ws,           // (B,)      [coherent]
    float* __restrict__ var_ws,            // (B, NCHUNK) [coherent]
    int n_batches, int batch_size, int pool_size, int dc, int n,
    long long it_start, long long iterations,
    float visibility, float gravity, float neg_gravity, float norm_scale,
    float penalize_factor, float perturbation_lower_bound,
    float base_perturbation,
    unsigned long long seed_suggest, unsigned long long seed_update,
    float amp2, float mean_c, int acq, float coef, float best_value,
    float tr_radius) {
  __shared__ float scale[MAX_POOL];
  __shared__ float red[8];
  __shared__ float s_scalar;
  __shared__ float xq_lds[512];
  __shared__ float pool_lds[POOL_LDS_CAP];

  const int tid = threadIdx.x;
  const int G = gridDim.x;
  const int pool_elems = pool_size * dc;
  const bool stage_pool = pool_elems <= POOL_LDS_CAP;
  auto fsum = [](float a, float c) { return a + c; };
  auto fmin_ = [](float a, float c) { return fminf(a, c); };
  auto fmax_ = [](float a, float c) { return fmaxf(a, c); };

  // Per-phase cycle profiling (workgroup 0's view, including its
  // barrier-wait slack): accumulated into iter_ptr[2..9] as
  // A, barA, B, barB, B2, barB2, C, barC. Reads are a handful of
  // s_memtime per iteration on one lane — no measurable perturbation.
  const bool profwg = (blockIdx.x == 0 && threadIdx.x == 0);
  unsigned long long tmark = profwg ? wall_clock64() : 0;
#define VZ_SWEEP_PROF(slot)                                        \
  if (profwg) {                                                    \
    const unsigned long long tnow = wall_clock64();                \
    iter_ptr[2 + (slot)] += tnow - tmark;                          \
    tmark = tnow;                                                  \
  }

  for (long long it = it_start; it < it_start + iterations; ++it) {
    const unsigned long long offset = (unsigned long long)it;
    const int batch_start =
        (int)(offset % (unsigned long long)n_batches) * batch_size;
    if (profwg) tmark = wall_clock64();

    // ---- Phase A: suggest + k-vec (workgroup owns candidate b) ----
    for (int b = blockIdx.x; b < batch_size; b += G) {
      if (stage_pool) {
        for (int e = tid; e < pool_elems; e += BLOCK) {
          pool_lds[e] = cload(pool_cont + e);
        }
        __syncthreads();
      }
      const int me = batch_start + b;
      const float my_reward = cload(rewards + me);
      const float my_pert = cload(perturbations + me);
      for (int p = tid; p < pool_size; p += BLOCK) {
        float d2 = 0.0f;
        if (stage_pool) {
          for (int j = 0; j < dc; ++j) {
            const float diff = pool_lds[me * dc + j] - pool_lds[p * dc + j];
            d2 = fmaf(diff, diff, d2);
          }
        } else {
          for (int j = 0; j < dc; ++j) {
            const float diff = cload(pool_cont + me * dc + j) -
                               cload(pool_cont + p * dc + j);
            d2 = fmaf(diff, diff, d2);
          }
        }
        const float reward_p = cload(rewards + p);
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
        if (stage_pool) {
          for (int p = 0; p < pool_size; ++p) {
            moved = fmaf(scale[p], pool_lds[p * dc + j], moved);
          }
        } else {
          for (int p = 0; p < pool_size; ++p) {
            moved = fmaf(scale[p], cload(pool_cont + p * dc + j), moved);
          }
        }
        const float mine = stage_pool ? pool_lds[me * dc + j]
                                      : cload(pool_cont + me * dc + j);
        moved = mine + (moved - mine * scale_sum);
        float noise = vz_rng_laplace(seed_suggest, offset,
                                     (unsigned)(b * dc + j));
        noise = (noise >= 0.0f) ? 1.0f : -1.0f;  // q == 1
        const float cand = moved + noise * my_pert;
        out_cont[b * dc + j] = cand;
        xq_lds[j] = cand;
      }
      __syncthreads();

      // k-vector + mu + trust-region distance for candidate b.
      float mu_acc = 0.0f;
      float min_linf = INFINITY;
      for (int row = tid; row < n; row += BLOCK) {
        const float* xr = x + (long)row * dc;
        float d2 = 0.0f, linf = 0.0f;
        for (int j = 0; j < dc; ++j) {
          const float diff = xq_lds[j] - xr[j];
          const float z = diff * inv_ls[j];
          d2 = fmaf(z, z, d2);
          linf = fmaxf(linf, fabsf(diff));  // continuous-only
        }
        const float kv = amp2 * matern52_of_d2(d2);
        cstore(k_ws + (long)b * n + row, kv);
        mu_acc = fmaf(kv, alpha[row], mu_acc);
        min_linf = fminf(min_linf, linf);
      }
      float mu = block_reduce(mu_acc, red, fsum, 0.0f);
      if (tid == 0) cstore(mu_ws + b, mu);
      __syncthreads();
      float dist = block_reduce(min_linf, red, fmin_, INFINITY);
      if (tid == 0) cstore(dist_ws + b, dist);
      __syncthreads();
    }
    VZ_SWEEP_PROF(0)
    grid_sync(barrier_buf);
    VZ_SWEEP_PROF(1)

    // ---- Phase B: K^-1 quadform (64x64 tiles, Kinv read ONCE) ----
    // Shared template with the standalone chunked scorer
    // (vz_quadform_tile in common.h): identical float-op order keeps
    // the megakernel bit-identical to the hipGraph path. var_ws layout
    // is (B, T+1): T tile partials + the reduced quadform.
    {
      const int tiles_n = (n + VZ_QF_TILE - 1) / VZ_QF_TILE;
      const int total_tiles = tiles_n * tiles_n;
      float* k_i_lds = pool_lds;                       // 2048 floats
      float* k_j_lds = pool_lds + VZ_QF_QMAX * VZ_QF_TILE;
      for (int tile = blockIdx.x; tile < total_tiles; tile += G) {
        const int t = tile;
        vz_quadform_tile(
            kinv, batch_size, n, t, tiles_n, k_i_lds, k_j_lds,
            [&](long idx) { return cload(k_ws + idx); },
            [&](int q, float v) {
              cstore(var_ws + (long)q * (total_tiles + 1) + t, v);
            },
            blockIdx.x == 0 ? iter_ptr + 10 : nullptr);
      }
    }
    VZ_SWEEP_PROF(2)
    grid_sync(barrier_buf);
    VZ_SWEEP_PROF(3)

    // ---- Phase B2: reduce tile partials (order == the standalone
    // ps_reduce_parts_kernel) ----
    {
      const int tiles_n = (n + VZ_QF_TILE - 1) / VZ_QF_TILE;
      const int total_tiles = tiles_n * tiles_n;
      for (int q = blockIdx.x; q < batch_size; q += G) {
        float sacc = 0.0f;
        for (int w = tid; w < total_tiles; w += BLOCK) {
          sacc += cload(var_ws + (long)q * (total_tiles + 1) + w);
        }
        float total = block_reduce(sacc, red, fsum, 0.0f);
        if (tid == 0) {
          cstore(var_ws + (long)q * (total_tiles + 1) + total_tiles,
                 total);
        }
        __syncthreads();
      }
    }
    VZ_SWEEP_PROF(4)
    grid_sync(barrier_buf);
    VZ_SWEEP_PROF(5)

    // ---- Phase C: finalize scores (inline) + update ----
    for (int i = blockIdx.x; i < batch_size; i += G) {
      // Every workgroup recomputes all B scores from the coherent
      // partials (cheap, deterministic, removes a barrier).
      float local_max = -INFINITY;
      float my_score = -INFINITY;
      const int tiles_n_c = (n + VZ_QF_TILE - 1) / VZ_QF_TILE;
      const int total_tiles_c = tiles_n_c * tiles_n_c;
      for (int t = tid; t < batch_size; t += BLOCK) {
        const float quad =
            cload(var_ws + (long)t * (total_tiles_c + 1) + total_tiles_c);
        float var = fmaxf(amp2 - quad, 1e-12f);
        const float sd = sqrtf(var);
        const float mu = cload(mu_ws + t) + mean_c;
        float score = acq_score(acq, mu, sd, coef, best_value);
        const float dist = cload(dist_ws + t);
        if (tr_radius > 0.0f && tr_radius <= 0.5f && dist > tr_radius) {
          score = -1e4f - dist;
        }
        local_max = fmaxf(local_max, score);
        if (t == i) my_score = score;
      }
      float batch_max = block_reduce(local_max, red, fmax_, -INFINITY);
      // my_score lives in the thread with tid == i%BLOCK (B <= BLOCK):
      // broadcast through LDS.
      if (tid == (i % BLOCK) && i < BLOCK) red[6] = my_score;
      __syncthreads();
      const float new_r = red[6];
      if (tid == 0) {
        s_scalar = fmaxf(cload(best_reward), batch_max);
        if (i == 0) cstore(best_reward, s_scalar);
      }
      __syncthreads();
      const float new_best = s_scalar;
      const int me = batch_start + i;
      const float old_r = cload(rewards + me);
      const bool improved = new_r > old_r;
      const float old_pert = cload(perturbations + me);
      float pert = improved ? old_pert : old_pert * penalize_factor;
      float reward = improved ? new_r : old_r;
      const bool dead = (pert < perturbation_lower_bound) &&
                        (reward != new_best);
      if (improved) {
        for (int j = tid; j < dc; j += BLOCK) {
          cstore(pool_cont + me * dc + j, out_cont[i * dc + j]);
        }
      }
      if (dead) {
        for (int j = tid; j < dc; j += BLOCK) {
          cstore(pool_cont + me * dc + j,
                 vz_rng_uniform(seed_update, offset ^ 0x5151ull,
                                (unsigned)(me * dc + j)));
        }
        reward = -INFINITY;
        pert = base_perturbation;
      }
      if (tid == 0) {
        cstore(rewards + me, reward);
        cstore(perturbations + me, pert);
      }
      __syncthreads();
    }
    VZ_SWEEP_PROF(6)
    grid_sync(barrier_buf);
    VZ_SWEEP_PROF(7)
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
    float tr_radius, hipStream_t stream) {
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
  const int tiles_n = (n + VZ_QF_TILE - 1) / VZ_QF_TILE;
  // Grid size is a pure throughput knob (every phase is a grid-stride
  // loop and partial-sum layout is grid-independent, so results are
  // bit-identical at any size). Measured at the headline shape
  // (N=1000, batch 25): tiles_n^2=256 WGs -> 61 us/iter but 64-128
  // WGs -> ~42 us/iter — the sense-reversing barrier's cost grows
  // with arrival count, so cap the default at 128 and floor at 64
  // (tiles_n^2 starves phase A at small N: 4 WGs at N=125 ran at
  // 107 us/iter). Override with VIZIER_AMD_SWEEP_GRID.
  int grid = tiles_n * tiles_n;                   // fills phase B
  if (grid > 128) grid = 128;
  if (grid < 64) grid = 64;
  static int grid_env = -1;
  if (grid_env < 0) {
    const char* env = getenv("VIZIER_AMD_SWEEP_GRID");
    grid_env = (env != nullptr) ? atoi(env) : 0;
  }
  if (grid_env > 0) grid = grid_env;
  const int max_grid = blocks_per_cu * num_cu;
  if (grid > max_grid) grid = max_grid;
  if (grid < 1) grid = 1;

  void* args[] = {
      &pool_cont, &rewards, &perturbations, &best_reward, &iter_ptr,
      &barrier_buf, &x, &inv_ls, &alpha, &kinv, &out_cont, &k_ws,
      &mu_ws, &dist_ws, &var_ws, &n_batches, &batch_size, &pool_size,
      &dc, &n, &it_start, &iterations, &visibility, &gravity,
      &neg_gravity, &norm_scale, &penalize_factor,
      &perturbation_lower_bound, &base_perturbation, &seed_suggest,
      &seed_update, &amp2, &mean_c, &acq, &coef, &best_value,
      &tr_radius};
  err = hipLaunchCooperativeKernel((const void*)eagle_sweep_kernel,
                                   dim3(grid), dim3(BLOCK), args, 0,
                                   stream);
  if (err != hipSuccess) return -(int)err;
  return grid;
}
