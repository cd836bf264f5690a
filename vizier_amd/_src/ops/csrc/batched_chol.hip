// Batched Cholesky + forward substitution for the NLL line search
// (gfx950). MAGMA's batched spotf2 panels cost ~54 ms per suggest at
// the headline shape (16 x 1000 x 1000 per L-BFGS iteration,
// profiles/sweep_kernels_r2.txt) in ~5000 tiny kernel launches, and
// the (R, N, 1) triangular solves dispatch as SERIAL rocblas trsv
// calls (~170 us each). These kernels run one WORKGROUP per matrix:
// a right-looking blocked factorization (NB=32 panels staged in LDS,
// register-held L21 rows for the trailing update) and a
// wave-synchronous blocked forward substitution. Forward-only: the
// autograd value_and_grad path keeps torch's factorization; the
// line-search (torch.no_grad) calls these.

#include <hip/hip_runtime.h>

#include <cstdlib>

#include "common.h"

#define CB 256
#define NB 32

extern "C" __global__ __launch_bounds__(CB) void
batched_potrf_kernel(float* __restrict__ A,    // (R, N, N) in-place L
                     int* __restrict__ info,   // (R,)
                     int r_count, int n) {
  __shared__ float diag[NB][NB + 1];
  __shared__ float jpanel[64][NB];
  __shared__ int s_info;
  const int r = blockIdx.x;
  if (r >= r_count) return;
  float* M = A + (long)r * n * n;
  const int tid = threadIdx.x;
  if (tid == 0) s_info = 0;
  __syncthreads();

  for (int k0 = 0; k0 < n; k0 += NB) {
    const int nb = min(NB, n - k0);
    for (int e = tid; e < nb * nb; e += CB) {
      diag[e / nb][e % nb] = M[(long)(k0 + e / nb) * n + k0 + e % nb];
    }
    __syncthreads();
    // Unblocked factor of the diagonal block in LDS.
    for (int j = 0; j < nb; ++j) {
      if (tid == 0) {
        const float v = diag[j][j];
        if (v > 0.0f) {
          diag[j][j] = sqrtf(v);
        } else {
          diag[j][j] = 1.0f;  // keep going; info marks the failure
          if (s_info == 0) s_info = k0 + j + 1;
        }
      }
      __syncthreads();
      const float dj = diag[j][j];
      for (int i = j + 1 + tid; i < nb; i += CB) {
        diag[i][j] /= dj;
      }
      __syncthreads();
      const int rem = nb - j - 1;
      for (int e = tid; e < rem * rem; e += CB) {
        const int i = j + 1 + e / rem;
        const int c = j + 1 + e % rem;
        if (c <= i) diag[i][c] -= diag[i][j] * diag[c][j];
      }
      __syncthreads();
    }
    for (int e = tid; e < nb * nb; e += CB) {
      const int i = e / nb, c = e % nb;
      M[(long)(k0 + i) * n + k0 + c] = (c <= i) ? diag[i][c] : 0.0f;
    }
    __syncthreads();

    // Column panel: L21 = A21 * L11^-T, one row per thread.
    for (int i = k0 + nb + tid; i < n; i += CB) {
      float v[NB];
#pragma unroll
      for (int j = 0; j < NB; ++j) {
        if (j < nb) {
          float xv = M[(long)i * n + k0 + j];
#pragma unroll
          for (int p = 0; p < NB; ++p) {
            if (p < j) xv -= v[p] * diag[j][p];
          }
          v[j] = xv / diag[j][j];
        }
      }
#pragma unroll
      for (int j = 0; j < NB; ++j) {
        if (j < nb) M[(long)i * n + k0 + j] = v[j];
      }
    }
    __syncthreads();

    // Trailing update (lower triangle only): A22 -= L21 L21^T.
    for (int jb = k0 + nb; jb < n; jb += 64) {
      const int jl = min(64, n - jb);
      for (int e = tid; e < jl * nb; e += CB) {
        jpanel[e / nb][e % nb] = M[(long)(jb + e / nb) * n + k0 + e % nb];
      }
      __syncthreads();
      for (int i = jb + tid; i < n; i += CB) {
        float row[NB];
#pragma unroll
        for (int p = 0; p < NB; ++p) {
          row[p] = (p < nb) ? M[(long)i * n + k0 + p] : 0.0f;
        }
        for (int j = 0; j < jl; ++j) {
          if (jb + j > i) break;
          float acc = 0.0f;
#pragma unroll
          for (int p = 0; p < NB; ++p) {
            acc = fmaf(row[p], jpanel[j][p], acc);
          }
          M[(long)i * n + jb + j] -= acc;
        }
      }
      __syncthreads();
    }
  }
  if (tid == 0) info[r] = s_info;
}

// Blocked forward substitution: solves L z = b in place for (R, N)
// right-hand sides. Wave 0 runs the wave-synchronous 32-panel solve
// (shfl broadcasts, no block syncs inside the panel); all waves apply
// the trailing update.
extern "C" __global__ __launch_bounds__(CB) void
batched_trsv_lower_kernel(const float* __restrict__ L,  // (R, N, N)
                          float* __restrict__ b,        // (R, N)
                          int r_count, int n) {
  __shared__ float zseg[NB];
  __shared__ float rows_lds[CB][NB + 1];
  const int r = blockIdx.x;
  if (r >= r_count) return;
  const float* M = L + (long)r * n * n;
  float* rhs = b + (long)r * n;
  const int tid = threadIdx.x;
  const int lane = tid % WAVE_SIZE;

  for (int k0 = 0; k0 < n; k0 += NB) {
    const int nb = min(NB, n - k0);
    if (tid < WAVE_SIZE) {
      // Lane l preloads ITS row of the 32x32 diagonal block into
      // registers up front (one parallel burst), so the serial
      // 32-step substitution touches only registers + shfl. The
      // previous form read M inside the chain — 32 dependent L2
      // round trips per panel, ~800 us per 1000-row solve
      // (profiles/fit_kernels_headline3.txt).
      float row[NB];
#pragma unroll
      for (int c = 0; c < NB; ++c) {
        row[c] = (lane < nb && c < nb && c <= lane)
            ? M[(long)(k0 + lane) * n + k0 + c] : 0.0f;
      }
      float myv = (lane < nb) ? rhs[k0 + lane] : 0.0f;
      for (int j = 0; j < nb; ++j) {
        float zj = 0.0f;
        if (j == lane) {
          zj = myv / row[j];
          zseg[j] = zj;
        }
        zj = __shfl(zj, j, WAVE_SIZE);
        if (lane > j && lane < nb) {
          myv = fmaf(-row[j], zj, myv);
        }
      }
      if (lane < nb) rhs[k0 + lane] = zseg[lane];
    }
    __syncthreads();
    // Row blocks staged through LDS for coalesced L-column reads.
    for (int i0 = k0 + nb; i0 < n; i0 += CB) {
      const int blk = min(CB, n - i0);
      for (int e = tid; e < blk * nb; e += CB) {
        rows_lds[e / nb][e % nb] =
            M[(long)(i0 + e / nb) * n + k0 + e % nb];
      }
      __syncthreads();
      const int i = i0 + tid;
      if (i < n) {
        float acc = rhs[i];
#pragma unroll
        for (int j = 0; j < NB; ++j) {
          if (j < nb) acc = fmaf(-rows_lds[tid][j], zseg[j], acc);
        }
        rhs[i] = acc;
      }
      __syncthreads();
    }
  }
}

// -- v2: panel-swept factorization --------------------------------------
//
// v1 (one workgroup per matrix for the WHOLE factorization) measured
// slower than MAGMA in context: at batch 16 only 16 CUs work and the
// trailing updates serialize inside each workgroup. v2 keeps the
// panel factor per-matrix (small) but launches the trailing update as
// a (R x column-tiles) grid per panel — 2 launches per panel (~64
// total) with the O(N^2) trailing work spread over the chip.

extern "C" __global__ __launch_bounds__(CB) void
batched_potrf_panel_kernel(float* __restrict__ A, int* __restrict__ info,
                           int r_count, int n, int k0) {
  __shared__ float diag[NB][NB + 1];
  __shared__ float rows_lds[CB][NB + 1];
  const int r = blockIdx.x;
  if (r >= r_count) return;
  // Grid (R, row-tiles): each workgroup REDUNDANTLY factors the 32x32
  // diagonal block (deterministic, cheap — wave-synchronous with shfl
  // broadcasts) then solves its OWN CB-row slice of the column panel
  // L21 = A21 L11^-T. The r2 single-workgroup-per-matrix version put
  // ~70 us x 32 panels on 3 CUs at the headline shape (R=3, N=1000,
  // profiles/fit_kernels_headline.txt); slicing rows across blockIdx.y
  // lets the panel use the chip, and the panel block is staged through
  // LDS so global accesses are COALESCED (one thread per row reads a
  // 128-byte segment strided n*4 from its neighbor's — measured ~52 us
  // per launch before staging, launch floor is ~4.5 us).
  const int row0 = k0 + NB + blockIdx.y * CB;
  const bool has_rows = row0 < n;
  if (blockIdx.y > 0 && !has_rows) return;
  float* M = A + (long)r * n * n;
  const int tid = threadIdx.x;
  const int nb = min(NB, n - k0);
  if (tid < WAVE_SIZE) {
    const int lane = tid;
    float row[NB];
#pragma unroll
    for (int c = 0; c < NB; ++c) {
      row[c] = (lane < nb && c < nb)
          ? M[(long)(k0 + lane) * n + k0 + c] : 0.0f;
    }
#pragma unroll
    for (int j = 0; j < NB; ++j) {
      if (j >= nb) break;
      const float piv = __shfl(row[j], j, WAVE_SIZE);
      float d;
      if (piv > 0.0f) {
        d = sqrtf(piv);
      } else {
        d = 1.0f;
        if (blockIdx.y == 0 && lane == 0 && info[r] == 0) {
          info[r] = k0 + j + 1;
        }
      }
      if (lane == j) row[j] = d;
      if (lane > j) row[j] /= d;
#pragma unroll
      for (int c = 0; c < NB; ++c) {
        if (c > j && c < nb) {
          const float lcj = __shfl(row[j], c, WAVE_SIZE);
          if (lane >= c) row[c] -= row[j] * lcj;
        }
      }
    }
#pragma unroll
    for (int c = 0; c < NB; ++c) {
      if (lane < nb && c < nb) diag[lane][c] = row[c];
    }
  }
  __syncthreads();
  if (blockIdx.y == 0) {
    for (int e = tid; e < nb * nb; e += CB) {
      const int i = e / nb, c = e % nb;
      M[(long)(k0 + i) * n + k0 + c] = (c <= i) ? diag[i][c] : 0.0f;
    }
  }
  const int tile_rows = has_rows ? min(CB, n - row0) : 0;
  for (int e = tid; e < tile_rows * nb; e += CB) {
    rows_lds[e / nb][e % nb] = M[(long)(row0 + e / nb) * n + k0 + e % nb];
  }
  __syncthreads();
  if (tid < tile_rows) {  // one row per thread in this tile
    float v[NB];
#pragma unroll
    for (int j = 0; j < NB; ++j) {
      if (j < nb) {
        float xv = rows_lds[tid][j];
#pragma unroll
        for (int p = 0; p < NB; ++p) {
          if (p < j) xv -= v[p] * diag[j][p];
        }
        v[j] = xv / diag[j][j];
      }
    }
#pragma unroll
    for (int j = 0; j < NB; ++j) {
      if (j < nb) rows_lds[tid][j] = v[j];
    }
  }
  __syncthreads();
  for (int e = tid; e < tile_rows * nb; e += CB) {
    M[(long)(row0 + e / nb) * n + k0 + e % nb] = rows_lds[e / nb][e % nb];
  }
}

extern "C" __global__ __launch_bounds__(CB) void
batched_potrf_trailing_kernel(float* __restrict__ A, int r_count, int n,
                              int k0) {
  __shared__ float jpanel[64][NB];
  __shared__ float rows_lds[CB][NB + 1];
  const int r = blockIdx.x;
  if (r >= r_count) return;
  float* M = A + (long)r * n * n;
  const int tid = threadIdx.x;
  const int nb = min(NB, n - k0);
  const int jb = k0 + nb + blockIdx.y * 64;
  if (jb >= n) return;
  const int jl = min(64, n - jb);
  for (int e = tid; e < jl * nb; e += CB) {
    jpanel[e / nb][e % nb] = M[(long)(jb + e / nb) * n + k0 + e % nb];
  }
  __syncthreads();
  // Row blocks staged through LDS for coalesced panel reads (the
  // per-thread row loads are 4-KB-strided otherwise).
  for (int i0 = jb; i0 < n; i0 += CB) {
    const int blk = min(CB, n - i0);
    for (int e = tid; e < blk * nb; e += CB) {
      rows_lds[e / nb][e % nb] =
          M[(long)(i0 + e / nb) * n + k0 + e % nb];
    }
    __syncthreads();
    const int i = i0 + tid;
    if (i < n) {
      float row[NB];
#pragma unroll
      for (int p = 0; p < NB; ++p) {
        row[p] = (p < nb) ? rows_lds[tid][p] : 0.0f;
      }
      for (int j = 0; j < jl; ++j) {
        if (jb + j > i) break;
        float acc = 0.0f;
#pragma unroll
        for (int p = 0; p < NB; ++p) {
          acc = fmaf(row[p], jpanel[j][p], acc);
        }
        M[(long)i * n + jb + j] -= acc;
      }
    }
    __syncthreads();
  }
}

// -- v3: persistent cooperative factorization (OPT-IN, measured slower)
//
// One cooperative launch for all panels with eagle_sweep-style grid
// barriers, pair (r, t) owning 64-row tile t. MEASURED RESULT at the
// headline shape: warm refit 132.9 -> 240.9 ms — SLOWER than v2's 64
// dependent launches. Root cause: a right-looking factorization
// rewrites the whole trailing matrix every round, and cross-workgroup
// coherence inside one kernel forces those reads+writes through
// agent-scope (memory-side, uncached) accesses — ~1.3 GB of L2-bypass
// traffic per N=1000 factorization, which costs more than the launch
// turnarounds it saves. (eagle_sweep wins with this pattern because
// its bulk data — K^-1, x, alpha — is READ-ONLY and stays L2-cached;
// only small buffers cross workgroups.) Kept opt-in
// (VIZIER_AMD_COOP_CHOL=1) as a documented experiment.

__device__ __forceinline__ float cchol_load(const float* p) {
  return __hip_atomic_load(p, __ATOMIC_RELAXED, __HIP_MEMORY_SCOPE_AGENT);
}
__device__ __forceinline__ void cchol_store(float* p, float v) {
  __hip_atomic_store(p, v, __ATOMIC_RELAXED, __HIP_MEMORY_SCOPE_AGENT);
}
__device__ __forceinline__ int cchol_iload(const int* p) {
  return __hip_atomic_load(p, __ATOMIC_RELAXED, __HIP_MEMORY_SCOPE_AGENT);
}
__device__ __forceinline__ void cchol_istore(int* p, int v) {
  __hip_atomic_store(p, v, __ATOMIC_RELAXED, __HIP_MEMORY_SCOPE_AGENT);
}

__device__ __forceinline__ void chol_grid_sync(unsigned int* bar) {
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
      __builtin_amdgcn_s_waitcnt(0);
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

#define RB 64  // rows per ownership tile

extern "C" __global__ __launch_bounds__(CB) void
batched_potrf_coop_kernel(float* __restrict__ A, int* __restrict__ info,
                          unsigned int* __restrict__ bar,
                          int r_count, int n) {
  __shared__ float diag[NB][NB + 1];
  __shared__ float myl[RB][NB];
  __shared__ float jpanel[RB][NB];
  const int tid = threadIdx.x;
  const int tiles = (n + RB - 1) / RB;
  const int npairs = r_count * tiles;

  for (int k0 = 0; k0 < n; k0 += NB) {
    const int nb = min(NB, n - k0);
    // Phase A/B: redundant diagonal factor + owned panel rows. The
    // factored diagonal is NOT written back here — every pair reads
    // the ORIGINAL diag values, so a writeback would race with other
    // pairs' loads. The owner writes it in phase C (whose trailing
    // updates touch disjoint rows) after a cheap redundant re-factor.
    for (int pair = blockIdx.x; pair < npairs; pair += gridDim.x) {
      const int r = pair / tiles, t = pair % tiles;
      const int tile_lo = t * RB;
      const int tile_hi = min(tile_lo + RB, n);
      if (tile_hi <= k0 + nb) continue;  // no panel rows in this tile
      float* M = A + (long)r * n * n;
      if (tid < WAVE_SIZE) {
        const int lane = tid;
        float row[NB];
#pragma unroll
        for (int c = 0; c < NB; ++c) {
          row[c] = (lane < nb && c < nb)
              ? cchol_load(&M[(long)(k0 + lane) * n + k0 + c]) : 0.0f;
        }
#pragma unroll
        for (int j = 0; j < NB; ++j) {
          if (j >= nb) break;
          const float piv = __shfl(row[j], j, WAVE_SIZE);
          const float d = (piv > 0.0f) ? sqrtf(piv) : 1.0f;
          if (lane == j) row[j] = d;
          if (lane > j) row[j] /= d;
#pragma unroll
          for (int c = 0; c < NB; ++c) {
            if (c > j && c < nb) {
              const float lcj = __shfl(row[j], c, WAVE_SIZE);
              if (lane >= c) row[c] -= row[j] * lcj;
            }
          }
        }
#pragma unroll
        for (int c = 0; c < NB; ++c) {
          if (lane < nb && c < nb) diag[lane][c] = row[c];
        }
      }
      __syncthreads();
      const int row_lo = max(tile_lo, k0 + nb);
      const int i = row_lo + tid;  // one row per thread (<= 64 rows)
      if (i < tile_hi) {
        float v[NB];
#pragma unroll
        for (int j = 0; j < NB; ++j) {
          if (j < nb) {
            float xv = cchol_load(&M[(long)i * n + k0 + j]);
#pragma unroll
            for (int p = 0; p < NB; ++p) {
              if (p < j) xv -= v[p] * diag[j][p];
            }
            v[j] = xv / diag[j][j];
          }
        }
#pragma unroll
        for (int j = 0; j < NB; ++j) {
          if (j < nb) cchol_store(&M[(long)i * n + k0 + j], v[j]);
        }
      }
      __syncthreads();  // diag LDS reused by the next pair
    }
    chol_grid_sync(bar);

    // Phase C: owner writes the factored diagonal (re-factored from
    // the still-original values — trailing writes touch disjoint
    // rows), then the trailing update of owned rows (identical fma
    // chains to batched_potrf_trailing_kernel).
    for (int pair = blockIdx.x; pair < npairs; pair += gridDim.x) {
      const int r = pair / tiles, t = pair % tiles;
      const int tile_lo = t * RB;
      const int tile_hi = min(tile_lo + RB, n);
      float* M = A + (long)r * n * n;
      if (t == k0 / RB) {
        if (tid < WAVE_SIZE) {
          const int lane = tid;
          float row[NB];
#pragma unroll
          for (int c = 0; c < NB; ++c) {
            row[c] = (lane < nb && c < nb)
                ? cchol_load(&M[(long)(k0 + lane) * n + k0 + c]) : 0.0f;
          }
#pragma unroll
          for (int j = 0; j < NB; ++j) {
            if (j >= nb) break;
            const float piv = __shfl(row[j], j, WAVE_SIZE);
            float d;
            if (piv > 0.0f) {
              d = sqrtf(piv);
            } else {
              d = 1.0f;
              if (lane == 0 && cchol_iload(&info[r]) == 0) {
                cchol_istore(&info[r], k0 + j + 1);
              }
            }
            if (lane == j) row[j] = d;
            if (lane > j) row[j] /= d;
#pragma unroll
            for (int c = 0; c < NB; ++c) {
              if (c > j && c < nb) {
                const float lcj = __shfl(row[j], c, WAVE_SIZE);
                if (lane >= c) row[c] -= row[j] * lcj;
              }
            }
          }
#pragma unroll
          for (int c = 0; c < NB; ++c) {
            if (lane < nb && c < nb) diag[lane][c] = row[c];
          }
        }
        __syncthreads();
        for (int e = tid; e < nb * nb; e += CB) {
          const int i = e / nb, c = e % nb;
          cchol_store(&M[(long)(k0 + i) * n + k0 + c],
                      (c <= i) ? diag[i][c] : 0.0f);
        }
        __syncthreads();
      }
      const int row_lo = max(tile_lo, k0 + nb);
      if (row_lo >= tile_hi) continue;
      const int nrows = tile_hi - row_lo;
      for (int e = tid; e < nrows * nb; e += CB) {
        myl[e / nb][e % nb] =
            cchol_load(&M[(long)(row_lo + e / nb) * n + k0 + e % nb]);
      }
      __syncthreads();
      for (int jb = k0 + nb; jb < tile_hi; jb += RB) {
        const int jl = min(RB, n - jb);
        for (int e = tid; e < jl * nb; e += CB) {
          jpanel[e / nb][e % nb] =
              cchol_load(&M[(long)(jb + e / nb) * n + k0 + e % nb]);
        }
        __syncthreads();
        const int rl = tid >> 2, jq = tid & 3;
        const int i = row_lo + rl;
        if (rl < nrows && i >= jb) {
          for (int q = 0; q < 16; ++q) {
            const int j = jq * 16 + q;
            if (j < jl && jb + j <= i) {
              float acc = 0.0f;
#pragma unroll
              for (int p = 0; p < NB; ++p) {
                acc = fmaf(myl[rl][p], jpanel[j][p], acc);
              }
              float* dst = &M[(long)i * n + jb + j];
              cchol_store(dst, cchol_load(dst) - acc);
            }
          }
        }
        __syncthreads();
      }
    }
    chol_grid_sync(bar);
  }
}

extern "C" int launch_batched_potrf_coop(float* A, int* info,
                                         unsigned int* bar, int r, int n,
                                         hipStream_t stream) {
  int max_blocks = 0;
  hipError_t err = hipOccupancyMaxActiveBlocksPerMultiprocessor(
      &max_blocks, (const void*)batched_potrf_coop_kernel, CB, 0);
  if (err != hipSuccess || max_blocks < 1) return -1;
  hipDeviceProp_t prop;
  if (hipGetDeviceProperties(&prop, 0) != hipSuccess) return -1;
  const int tiles = (n + RB - 1) / RB;
  int grid = r * tiles;
  const int cap = max_blocks * prop.multiProcessorCount;
  if (grid > cap) grid = cap;
  if (grid < 1) grid = 1;
  void* args[] = {&A, &info, &bar, &r, &n};
  err = hipLaunchCooperativeKernel(
      (const void*)batched_potrf_coop_kernel, dim3(grid), dim3(CB),
      args, 0, stream);
  return err == hipSuccess ? 0 : -1;
}

// -- v4: fused round kernel (default) -----------------------------------
//
// ONE launch per 32-panel round instead of v2's two (panel + trailing),
// with NO cross-workgroup coherence traffic: round k's kernel never
// writes anything another workgroup of the same launch reads. The
// trick is write deferral + recomputation:
//   - the factored 32x32 diagonal stays OUT of A until the final
//     cleanup kernel (each consumer re-factors it from A's stable
//     values in-wave — ~1 us, deterministic);
//   - L21 panel rows are recomputed from A's original column values by
//     every workgroup that needs them (solve vs the LDS diagonal), and
//     only written back for round k-1 (whose columns nothing in round
//     k reads);
//   - the trailing update is in-place on disjoint (i, j >= jb) tiles.
// Per-element fma chains are IDENTICAL to v2, so v4 results are
// bit-identical to v2's. Measured: the A/B that motivated this is the
// ~44 us launch+L2-flush turnaround BETWEEN dependent kernels (potrf
// R=3 == R=12 == 2.85 ms on v2 — launch-bound, not work-bound).

__device__ __forceinline__ void wave_factor_diag(
    const float* __restrict__ M, int n, int k0, int nb,
    float diag_out[NB][NB + 1], int* fail_col /* -1 if ok */) {
  const int lane = threadIdx.x;
  float row[NB];
  int fc = -1;
#pragma unroll
  for (int c = 0; c < NB; ++c) {
    row[c] = (lane < nb && c < nb)
        ? M[(long)(k0 + lane) * n + k0 + c] : 0.0f;
  }
#pragma unroll
  for (int j = 0; j < NB; ++j) {
    if (j >= nb) break;
    const float piv = __shfl(row[j], j, WAVE_SIZE);
    float d;
    if (piv > 0.0f) {
      d = sqrtf(piv);
    } else {
      d = 1.0f;
      if (fc < 0) fc = k0 + j + 1;
    }
    if (lane == j) row[j] = d;
    if (lane > j) row[j] /= d;
#pragma unroll
    for (int c = 0; c < NB; ++c) {
      if (c > j && c < nb) {
        const float lcj = __shfl(row[j], c, WAVE_SIZE);
        if (lane >= c) row[c] -= row[j] * lcj;
      }
    }
  }
#pragma unroll
  for (int c = 0; c < NB; ++c) {
    if (lane < nb && c < nb) diag_out[lane][c] = row[c];
  }
  *fail_col = fc;
}

__device__ __forceinline__ void solve_panel_row(
    const float* __restrict__ M, int n, int i, int k0, int nb,
    const float diag[NB][NB + 1], float v[NB]) {
#pragma unroll
  for (int j = 0; j < NB; ++j) {
    if (j < nb) {
      float xv = M[(long)i * n + k0 + j];
#pragma unroll
      for (int p = 0; p < NB; ++p) {
        if (p < j) xv -= v[p] * diag[j][p];
      }
      v[j] = xv / diag[j][j];
    }
  }
}

extern "C" __global__ __launch_bounds__(CB) void
batched_potrf_round_kernel(float* __restrict__ A, int* __restrict__ info,
                           int r_count, int n, int k0) {
  __shared__ float pdiag[NB][NB + 1];  // factored diag of round k-1
  __shared__ float diag[NB][NB + 1];   // factored diag of round k
  __shared__ float jpanel[64][NB];     // L21^k rows of this j-tile
  const int r = blockIdx.x;
  if (r >= r_count) return;
  float* M = A + (long)r * n * n;
  const int tid = threadIdx.x;
  const int nb = min(NB, n - k0);
  const int k0p = k0 - NB;

  // Step 0: finalize round k-1 — write L21^{k-1} for this tile's rows
  // (columns k0p, which nothing in round k reads).
  if (k0p >= 0) {
    int fcp = -1;
    if (tid < WAVE_SIZE) {
      wave_factor_diag(M, n, k0p, NB, pdiag, &fcp);
    }
    __syncthreads();
    if (blockIdx.y == 0 && tid == 0 && fcp > 0 && info[r] == 0) {
      info[r] = fcp;  // unique writer per launch; launches are ordered
    }
    const int w0 = k0 + (int)blockIdx.y * 64;
    for (int i = w0 + tid; i < min(w0 + 64, n); i += CB) {
      float v[NB];
      solve_panel_row(M, n, i, k0p, NB, pdiag, v);
#pragma unroll
      for (int j = 0; j < NB; ++j) {
        M[(long)i * n + k0p + j] = v[j];
      }
    }
  }

  // Step 1: trailing update for this j-tile (jb >= k0+nb), using L21^k
  // recomputed from A's still-original panel columns.
  const int jb = k0 + nb + (int)blockIdx.y * 64;
  if (jb >= n || nb < NB) return;  // last round has no trailing
  int fc = -1;
  if (tid < WAVE_SIZE) {
    wave_factor_diag(M, n, k0, nb, diag, &fc);
  }
  __syncthreads();
  const int jl = min(64, n - jb);
  // Recompute L21 rows of this j-tile into LDS (identical solve chain).
  for (int j = tid; j < jl; j += CB) {
    float v[NB];
    solve_panel_row(M, n, jb + j, k0, nb, diag, v);
#pragma unroll
    for (int p = 0; p < NB; ++p) jpanel[j][p] = v[p];
  }
  __syncthreads();
  for (int i = jb + tid; i < n; i += CB) {
    float v[NB];
    solve_panel_row(M, n, i, k0, nb, diag, v);
    for (int j = 0; j < jl; ++j) {
      if (jb + j > i) break;
      float acc = 0.0f;
#pragma unroll
      for (int p = 0; p < NB; ++p) {
        acc = fmaf(v[p], jpanel[j][p], acc);
      }
      M[(long)i * n + jb + j] -= acc;
    }
  }
}

// Final cleanup: factor every diagonal block in place (their inputs
// are stable once the preceding trailing rounds ran), zero the upper
// part, and record a failure in the LAST block (earlier blocks were
// recorded by their successor round's step 0).
extern "C" __global__ __launch_bounds__(CB) void
batched_potrf_final_kernel(float* __restrict__ A, int* __restrict__ info,
                           int r_count, int n) {
  __shared__ float diag[NB][NB + 1];
  const int r = blockIdx.x;
  if (r >= r_count) return;
  float* M = A + (long)r * n * n;
  const int tid = threadIdx.x;
  const int k0 = (int)blockIdx.y * NB;
  const int nb = min(NB, n - k0);
  int fc = -1;
  if (tid < WAVE_SIZE) {
    wave_factor_diag(M, n, k0, nb, diag, &fc);
  }
  __syncthreads();
  if (tid == 0 && fc > 0 && k0 + NB >= n && info[r] == 0) {
    info[r] = fc;
  }
  for (int e = tid; e < nb * nb; e += CB) {
    const int i = e / nb, c = e % nb;
    M[(long)(k0 + i) * n + k0 + c] = (c <= i) ? diag[i][c] : 0.0f;
  }
}

// -- v5: persistent LEFT-looking cooperative factorization --------------
//
// Why a 5th design: the per-kernel cost of v2's 63 launches is NOT
// launch overhead (a 63-deep chain of trivial kernels runs at 4.4
// us/launch) and NOT uncoalesced access (LDS staging changed nothing)
// — it is the end-of-kernel release fence, which writes back and
// invalidates the per-XCD L2s (~40-65 us when megabytes are dirty;
// same effect eagle_sweep.hip documents for device-scope fences). The
// fix must therefore run the whole factorization in ONE kernel while
// keeping the bulk data CACHED — which a right-looking update cannot
// do (v3: the trailing matrix is cross-workgroup mutable).
//
// Left-looking solves it:
//   - the INPUT matrix A is never written (output is separate), so
//     its reads are normal cached loads;
//   - finalized L panels live in a PANEL-MAJOR scratch (32 floats per
//     row per panel, 128-byte aligned) so no cache line ever spans
//     two panels; a line is cstore'd once at finalization and only
//     normal-loaded AFTER that, so caches can never hold a stale
//     pre-write copy;
//   - only the factored 32x32 diagonal block crosses workgroups
//     mid-round, through a tiny agent-scope scratch.
// A plain kernel afterwards copies panel-major L into the dense
// output with the upper triangle zeroed.
//
// Round k, pair (r, t) owning 64-row tile t (one pair per workgroup,
// enforced by the launcher):
//   P1: rows_acc[i][c] = A[i, k0+c] - sum_{kb<k} L[i,:] L[k0+c,:]
//       (4 threads x 8 columns per row; LDS-staged L tiles);
//       the diagonal owner wave-factors rows_acc, records info, and
//       publishes the factored diagonal via agent-scope scratch.
//   barrier.
//   P2: every tile solves its own rows >= k0+32 against the diagonal
//       and finalizes them into the panel scratch.
//   barrier.

#define V5_RB 64

extern "C" __global__ __launch_bounds__(CB) void
batched_potrf_left_kernel(
    const float* __restrict__ A,     // (R, N, N) input, read-only
    float* __restrict__ panels,      // (R, ceil(N/32), N, 32) scratch
    float* __restrict__ dscratch,    // (R, 32, 32) diag broadcast
    int* __restrict__ info,          // (R,)
    unsigned int* __restrict__ bar,  // (2,) zeroed
    int r_count, int n) {
  __shared__ float rows_acc[V5_RB][NB + 1];
  __shared__ float lrows[V5_RB][NB];   // own-tile L rows, one kb panel
  __shared__ float ldiag[NB][NB + 1];  // diag-rows L, one kb panel
  __shared__ float fdiag[NB][NB + 1];  // factored diagonal this round
  const int tid = threadIdx.x;
  const int tiles = (n + V5_RB - 1) / V5_RB;
  const int r = blockIdx.x / tiles;
  const int t = blockIdx.x % tiles;
  const int tile_lo = t * V5_RB;
  const int tile_hi = min(tile_lo + V5_RB, n);
  const float* Ain = A + (long)r * n * n;
  float* P = panels + (long)r * ((n + NB - 1) / NB) * n * NB;
  float* ds = dscratch + (long)r * NB * NB;
  const int n_rounds = (n + NB - 1) / NB;

  for (int k = 0; k < n_rounds; ++k) {
    const int k0 = k * NB;
    const int nb = min(NB, n - k0);
    const int row_lo = tile_lo > k0 ? tile_lo : k0;
    const bool active = tile_hi > k0;
    const bool owner = (k0 >= tile_lo && k0 < tile_hi);

    if (active) {
      // P1: update own rows' panel columns (left-looking GEMM).
      const int nrows = tile_hi - row_lo;
      const int rl = tid >> 2, cq = (tid & 3) * 8;
      float acc[8];
      const int i = row_lo + rl;
      if (rl < nrows) {
#pragma unroll
        for (int c = 0; c < 8; ++c) {
          acc[c] = (cq + c < nb) ? Ain[(long)i * n + k0 + cq + c] : 0.0f;
        }
      }
      for (int kb = 0; kb < k; ++kb) {
        const float* Pk = P + (long)kb * n * NB;
        for (int e = tid; e < nrows * NB; e += CB) {
          lrows[e / NB][e % NB] = Pk[(long)(row_lo + e / NB) * NB
                                     + e % NB];
        }
        for (int e = tid; e < nb * NB; e += CB) {
          ldiag[e / NB][e % NB] = Pk[(long)(k0 + e / NB) * NB + e % NB];
        }
        __syncthreads();
        if (rl < nrows) {
#pragma unroll
          for (int c = 0; c < 8; ++c) {
            float a = acc[c];
            if (cq + c < nb) {
#pragma unroll
              for (int p = 0; p < NB; ++p) {
                a = fmaf(-lrows[rl][p], ldiag[cq + c][p], a);
              }
            }
            acc[c] = a;
          }
        }
        __syncthreads();
      }
      if (rl < nrows) {
#pragma unroll
        for (int c = 0; c < 8; ++c) {
          rows_acc[rl][cq + c] = acc[c];
        }
      }
      __syncthreads();

      if (owner) {
        // Wave-factor the updated diagonal block (rows k0..k0+nb are
        // inside this tile) and publish it.
        const int d0 = k0 - row_lo;
        int fc = -1;
        if (tid < WAVE_SIZE) {
          const int lane = tid;
          float row[NB];
#pragma unroll
          for (int c = 0; c < NB; ++c) {
            row[c] = (lane < nb && c < nb) ? rows_acc[d0 + lane][c]
                                           : 0.0f;
          }
#pragma unroll
          for (int j = 0; j < NB; ++j) {
            if (j >= nb) break;
            const float piv = __shfl(row[j], j, WAVE_SIZE);
            float d;
            if (piv > 0.0f) {
              d = sqrtf(piv);
            } else {
              d = 1.0f;
              if (fc < 0) fc = k0 + j + 1;
            }
            if (lane == j) row[j] = d;
            if (lane > j) row[j] /= d;
#pragma unroll
            for (int c = 0; c < NB; ++c) {
              if (c > j && c < nb) {
                const float lcj = __shfl(row[j], c, WAVE_SIZE);
                if (lane >= c) row[c] -= row[j] * lcj;
              }
            }
          }
#pragma unroll
          for (int c = 0; c < NB; ++c) {
            if (lane < nb && c < nb) fdiag[lane][c] = row[c];
          }
          if (lane == 0 && fc > 0 && cchol_iload(&info[r]) == 0) {
            cchol_istore(&info[r], fc);
          }
        }
        __syncthreads();
        // Publish diag: agent-scope scratch for other tiles + the
        // finalized panel rows for later rounds' GEMMs.
        float* Pk = P + (long)k * n * NB;
        for (int e = tid; e < nb * nb; e += CB) {
          const int ii = e / nb, c = e % nb;
          cchol_store(&ds[ii * NB + c], fdiag[ii][c]);
          cchol_store(&Pk[(long)(k0 + ii) * NB + c],
                      (c <= ii) ? fdiag[ii][c] : 0.0f);
        }
      }
    }
    chol_grid_sync(bar);

    // P2: solve own rows below the diagonal block and finalize them.
    if (active) {
      const int row_lo2 = tile_lo > k0 + nb ? tile_lo : k0 + nb;
      if (row_lo2 < tile_hi) {
        if (!owner) {
          for (int e = tid; e < nb * nb; e += CB) {
            fdiag[e / nb][e % nb] = cchol_load(&ds[(e / nb) * NB
                                                   + e % nb]);
          }
        }
        __syncthreads();
        const int base = row_lo2 - (tile_lo > k0 ? tile_lo : k0);
        const int i = row_lo2 + tid;
        float* Pk = P + (long)k * n * NB;
        if (i < tile_hi) {
          float v[NB];
#pragma unroll
          for (int j = 0; j < NB; ++j) {
            if (j < nb) {
              float xv = rows_acc[base + tid][j];
#pragma unroll
              for (int p = 0; p < NB; ++p) {
                if (p < j) xv -= v[p] * fdiag[j][p];
              }
              v[j] = xv / fdiag[j][j];
            } else {
              v[j] = 0.0f;
            }
          }
#pragma unroll
          for (int j = 0; j < NB; ++j) {
            cchol_store(&Pk[(long)i * NB + j], v[j]);
          }
        }
      }
      __syncthreads();
    }
    chol_grid_sync(bar);
  }
}

// Copies panel-major L into the dense (R, N, N) output, zeroing the
// upper triangle (and the never-written rows above each panel).
extern "C" __global__ __launch_bounds__(CB) void
potrf_copy_out_kernel(const float* __restrict__ panels,
                      float* __restrict__ L, int r_count, int n) {
  const int r = blockIdx.x;
  if (r >= r_count) return;
  const int pk = blockIdx.y;
  const int p0 = pk * NB;
  const float* Pk = panels + ((long)r * ((n + NB - 1) / NB) + pk)
                    * (long)n * NB;
  float* M = L + (long)r * n * n;
  for (int e = threadIdx.x; e < n * NB; e += CB) {
    const int i = e / NB, p = e % NB;
    const int j = p0 + p;
    if (j < n) {
      M[(long)i * n + j] = (j <= i) ? Pk[(long)i * NB + p] : 0.0f;
    }
  }
}

extern "C" int launch_batched_potrf_v5(const float* A, float* panels,
                                       float* dscratch, float* L,
                                       int* info, unsigned int* bar,
                                       int r, int n,
                                       hipStream_t stream) {
  int max_blocks = 0;
  hipError_t err = hipOccupancyMaxActiveBlocksPerMultiprocessor(
      &max_blocks, (const void*)batched_potrf_left_kernel, CB, 0);
  if (err != hipSuccess || max_blocks < 1) return -1;
  hipDeviceProp_t prop;
  if (hipGetDeviceProperties(&prop, 0) != hipSuccess) return -1;
  const int tiles = (n + V5_RB - 1) / V5_RB;
  const int grid = r * tiles;
  if (grid > max_blocks * prop.multiProcessorCount) return -1;
  void* args[] = {&A, &panels, &dscratch, &info, &bar, &r, &n};
  err = hipLaunchCooperativeKernel(
      (const void*)batched_potrf_left_kernel, dim3(grid), dim3(CB),
      args, 0, stream);
  if (err != hipSuccess) return -1;
  const int pk = (n + NB - 1) / NB;
  hipLaunchKernelGGL(potrf_copy_out_kernel, dim3(r, pk), dim3(CB), 0,
                     stream, panels, L, r, n);
  return 0;
}

extern "C" void launch_batched_potrf_v2(float* A, int* info, int r,
                                        int n, hipStream_t stream) {
  for (int k0 = 0; k0 < n; k0 += NB) {
    const int nb = (n - k0) < NB ? (n - k0) : NB;
    const int rows = n - (k0 + nb);
    const int rtiles = rows > 0 ? (rows + CB - 1) / CB : 1;
    hipLaunchKernelGGL(batched_potrf_panel_kernel, dim3(r, rtiles),
                       dim3(CB), 0, stream, A, info, r, n, k0);
    const int jtiles = (rows + 63) / 64;
    if (jtiles > 0) {
      hipLaunchKernelGGL(batched_potrf_trailing_kernel,
                         dim3(r, jtiles), dim3(CB), 0, stream, A, r, n,
                         k0);
    }
  }
}

extern "C" void launch_batched_potrf(float* A, int* info, int r, int n,
                                     hipStream_t stream) {
  // v4 measured SLOWER than v2 at the headline shape (3.63 vs 2.85 ms,
  // R=3 N=1000): the fused kernel's three inlined 32-wide solves blow
  // its register budget (VGPR 256 + 248 AGPRs, occupancy 1 wave/SIMD
  // vs v2's 4). Opt-in via VIZIER_AMD_CHOL_IMPL=v4 for experiments.
  static int use_v4 = -1;
  if (use_v4 < 0) {
    const char* env = std::getenv("VIZIER_AMD_CHOL_IMPL");
    use_v4 = (env != nullptr && env[0] == 'v' && env[1] == '4') ? 1 : 0;
  }
  if (!use_v4) {
    launch_batched_potrf_v2(A, info, r, n, stream);
    return;
  }
  for (int k0 = 0; k0 < n; k0 += NB) {
    const int nb = (n - k0) < NB ? (n - k0) : NB;
    const int step0_tiles = (n - k0 + 63) / 64;
    const int jtiles = (n - (k0 + nb) + 63) / 64;
    int tiles = step0_tiles > jtiles ? step0_tiles : jtiles;
    if (tiles < 1) tiles = 1;
    if (k0 == 0 && jtiles == 0) break;  // single block: final only
    hipLaunchKernelGGL(batched_potrf_round_kernel, dim3(r, tiles),
                       dim3(CB), 0, stream, A, info, r, n, k0);
  }
  const int blocks = (n + NB - 1) / NB;
  hipLaunchKernelGGL(batched_potrf_final_kernel, dim3(r, blocks),
                     dim3(CB), 0, stream, A, info, r, n);
}

extern "C" void launch_batched_trsv_lower(const float* L, float* b,
                                          int r, int n,
                                          hipStream_t stream) {
  hipLaunchKernelGGL(batched_trsv_lower_kernel, dim3(r), dim3(CB), 0,
                     stream, L, b, r, n);
}
