// Shared helpers for the vizier_amd CDNA4 (gfx950) kernels.
#pragma once

#include <hip/hip_runtime.h>

#define WAVE_SIZE 64

__device__ __forceinline__ float matern52_of_d2(float d2) {
  // k(r) = (1 + sqrt5*r + 5r^2/3) exp(-sqrt5*r), r = sqrt(d2)
  const float r = sqrtf(fmaxf(d2, 0.0f));
  const float sr = 2.2360679774997896f * r;  // sqrt(5) * r
  return (1.0f + sr + sr * sr * (1.0f / 3.0f)) * __expf(-sr);
}

// Wave-wide reductions over all 64 lanes.
__device__ __forceinline__ float wave_reduce_sum(float v) {
#pragma unroll
  for (int offset = WAVE_SIZE / 2; offset > 0; offset >>= 1) {
    v += __shfl_down(v, offset, WAVE_SIZE);
  }
  return v;
}

__device__ __forceinline__ float wave_reduce_min(float v) {
#pragma unroll
  for (int offset = WAVE_SIZE / 2; offset > 0; offset >>= 1) {
    v = fminf(v, __shfl_down(v, offset, WAVE_SIZE));
  }
  return v;
}

__device__ __forceinline__ float wave_reduce_max(float v) {
#pragma unroll
  for (int offset = WAVE_SIZE / 2; offset > 0; offset >>= 1) {
    v = fmaxf(v, __shfl_down(v, offset, WAVE_SIZE));
  }
  return v;
}

// Block-level reduction: 256 threads = 4 waves -> LDS -> wave 0.
template <typename Op>
__device__ __forceinline__ float block_reduce(float v, float* lds4,
                                              Op op, float init) {
  const int lane = threadIdx.x & (WAVE_SIZE - 1);
  const int wave = threadIdx.x / WAVE_SIZE;
#pragma unroll
  for (int offset = WAVE_SIZE / 2; offset > 0; offset >>= 1) {
    v = op(v, __shfl_down(v, offset, WAVE_SIZE));
  }
  if (lane == 0) lds4[wave] = v;
  __syncthreads();
  if (wave == 0) {
    v = (lane < blockDim.x / WAVE_SIZE) ? lds4[lane] : init;
#pragma unroll
    for (int offset = 2; offset > 0; offset >>= 1) {
      v = op(v, __shfl_down(v, offset, WAVE_SIZE));
    }
  }
  return v;  // valid in wave 0 lane 0
}

// Counter-based RNG (splitmix64 -> uniform/laplace). Shared by the
// standalone Eagle kernels and the persistent sweep megakernel so the
// two paths draw IDENTICAL streams for the same (seed, offset, idx).
__device__ __forceinline__ unsigned long long vz_splitmix64(
    unsigned long long z) {
  z += 0x9e3779b97f4a7c15ull;
  z = (z ^ (z >> 30)) * 0xbf58476d1ce4e5b9ull;
  z = (z ^ (z >> 27)) * 0x94d049bb133111ebull;
  return z ^ (z >> 31);
}

__device__ __forceinline__ float vz_rng_uniform(unsigned long long seed,
                                                unsigned long long offset,
                                                unsigned int idx) {
  unsigned long long h = vz_splitmix64(seed ^ vz_splitmix64(offset ^ idx));
  return ((h >> 40) + 0.5f) * (1.0f / 16777216.0f);  // 24 bits -> (0,1)
}

__device__ __forceinline__ float vz_rng_laplace(unsigned long long seed,
                                                unsigned long long offset,
                                                unsigned int idx) {
  const float u = vz_rng_uniform(seed, offset, idx) - 0.5f;
  const float a = fminf(fabsf(u), 0.499999f);
  return (u >= 0.0f ? -1.0f : 1.0f) * log1pf(-2.0f * a);
}

__device__ __forceinline__ float vz_normal_cdf(float z) {
  return 0.5f * erfcf(-z * 0.70710678118654752f);
}
__device__ __forceinline__ float vz_normal_pdf(float z) {
  return 0.3989422804014327f * __expf(-0.5f * z * z);
}

// ---- Shared 64x64-tile K^-1 quadform --------------------------------------
//
// partial[q] contribution of tile (ti,tj) to quad[q] = k_q^T Kinv k_q,
// computed for ALL candidates at once so Kinv is read ONCE per scorer
// call. The per-candidate streaming variant reads Kinv b times per
// iteration (100 MB at N=1000, B=25) and measured 32.6 us of a 58 us
// Eagle iteration (profiles/sweep_kernels_r2.txt). Requires b <= 32
// and blockDim.x == 256. Used VERBATIM (same float-op order) by both
// the standalone chunked scorer and the persistent sweep megakernel so
// the two paths stay bit-identical.
#define VZ_QF_TILE 64
#define VZ_QF_QMAX 32

template <typename LoadK, typename StoreP>
__device__ __forceinline__ void vz_quadform_tile(
    const float* __restrict__ kinv, int b, int n, int tile, int tiles_n,
    float* k_i_lds, float* k_j_lds, LoadK loadk, StoreP store,
    unsigned long long* prof = nullptr) {
  const int tid = threadIdx.x;
  const bool profme = (prof != nullptr && tid == 0);
  unsigned long long tq0 = profme ? wall_clock64() : 0;
  const int ti = tile / tiles_n, tj = tile % tiles_n;
  const int i0 = ti * VZ_QF_TILE, j0 = tj * VZ_QF_TILE;
  const int ilen = min(VZ_QF_TILE, n - i0);
  const int jlen = min(VZ_QF_TILE, n - j0);
  for (int e = tid; e < VZ_QF_QMAX * VZ_QF_TILE; e += 256) {
    const int q = e / VZ_QF_TILE, c = e % VZ_QF_TILE;
    k_i_lds[e] = (q < b && c < ilen) ? loadk((long)q * n + i0 + c)
                                     : 0.0f;
    k_j_lds[e] = (q < b && c < jlen) ? loadk((long)q * n + j0 + c)
                                     : 0.0f;
  }
  __syncthreads();
  if (profme) {
    const unsigned long long tq1 = wall_clock64();
    prof[0] += tq1 - tq0;
    tq0 = tq1;
  }
  // Wave w owns candidates [8w, 8w+8); lane = tile column j. Kinv rows
  // are read coalesced across lanes (the 4 waves share each row
  // segment through L2).
  const int wave = tid / WAVE_SIZE;
  const int lane = tid % WAVE_SIZE;
  float acc[8];
#pragma unroll
  for (int qq = 0; qq < 8; ++qq) acc[qq] = 0.0f;
  if (lane < jlen) {
    if (ilen == VZ_QF_TILE) {
      // Full tile: batch the Kinv row loads 8 at a time so they are
      // all in flight together. The rolled one-load-per-iteration
      // form serializes 64 dependent L2 round-trips (~300 ns each),
      // which made phase B ~12 us per tile in the megakernel
      // (profiles: tools_sweep_phases.py). Same fma order per
      // candidate chain — results are bit-identical.
      for (int ib = 0; ib < VZ_QF_TILE; ib += 8) {
        float kvv[8];
#pragma unroll
        for (int u = 0; u < 8; ++u) {
          kvv[u] = kinv[(long)(i0 + ib + u) * n + j0 + lane];
        }
#pragma unroll
        for (int u = 0; u < 8; ++u) {
#pragma unroll
          for (int qq = 0; qq < 8; ++qq) {
            acc[qq] = fmaf(
                k_i_lds[(wave * 8 + qq) * VZ_QF_TILE + ib + u],
                kvv[u], acc[qq]);
          }
        }
      }
    } else {
      for (int i = 0; i < ilen; ++i) {
        const float kv = kinv[(long)(i0 + i) * n + j0 + lane];
#pragma unroll
        for (int qq = 0; qq < 8; ++qq) {
          acc[qq] = fmaf(k_i_lds[(wave * 8 + qq) * VZ_QF_TILE + i], kv,
                         acc[qq]);
        }
      }
    }
  }
#pragma unroll
  for (int qq = 0; qq < 8; ++qq) {
    const int q = wave * 8 + qq;
    float v = acc[qq] * k_j_lds[q * VZ_QF_TILE + lane];
    v = wave_reduce_sum(v);
    if (lane == 0 && q < b) store(q, v);
  }
  __syncthreads();
  if (profme) prof[1] += wall_clock64() - tq0;
}
