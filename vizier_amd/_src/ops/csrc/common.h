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
