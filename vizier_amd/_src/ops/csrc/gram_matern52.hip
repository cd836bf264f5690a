// Fused pairwise-distance + Matern-5/2 Gram kernel for gfx950.
//
// Computes K[i,j] = amp^2 * m52(||(x1_i - x2_j)/ls||) for x1 (N,D),
// x2 (M,D), both fp32 in [0,1]-ish range. LDS-tiled: each 16x16-thread
// workgroup computes a 16x16 output tile, staging both feature tiles
// through LDS in D-chunks (guide §5 pattern, distance instead of dot).
//
// Numeric spec: vizier_amd/_src/gp/matern.py::gram_matern52 (the CPU
// oracle used by tests/test_gpu_ops.py).

#include <hip/hip_runtime.h>
#include "common.h"

#define TILE 16
#define DCHUNK 32

extern "C" __global__ __launch_bounds__(TILE * TILE) void
gram_matern52_kernel(const float* __restrict__ x1,
                     const float* __restrict__ x2,
                     const float* __restrict__ inv_ls,  // 1/lengthscale (D)
                     float* __restrict__ out, int n, int m, int d,
                     float amp2, int sym /* x1 == x2: exploit symmetry */) {
  __shared__ float lds1[TILE][DCHUNK + 1];
  __shared__ float lds2[TILE][DCHUNK + 1];
  __shared__ float ldsl[DCHUNK];

  const int ty = threadIdx.x / TILE;   // row within tile
  const int tx = threadIdx.x % TILE;   // col within tile
  // XCD-aware swizzle of workgroup ids: consecutive ids land on the same
  // XCD's L2 so neighboring tiles share x1/x2 panels (guide T1).
  int nwg = gridDim.x;
  int wg = blockIdx.x;
  if (nwg % 8 == 0) {
    const int cpx = nwg / 8;
    wg = (wg % 8) * cpx + wg / 8;
  }
  const int tiles_m = (m + TILE - 1) / TILE;
  const int row0 = (wg / tiles_m) * TILE;
  const int col0 = (wg % tiles_m) * TILE;
  if (row0 >= n || col0 >= m) return;
  if (sym && col0 + TILE <= row0) return;  // strictly-lower tile: skip

  float acc = 0.0f;
  for (int d0 = 0; d0 < d; d0 += DCHUNK) {
    const int dc = min(DCHUNK, d - d0);
    // Cooperative staging: thread (ty, tx) loads strided columns.
    for (int c = tx; c < dc; c += TILE) {
      const int r1 = row0 + ty;
      const int r2 = col0 + ty;
      lds1[ty][c] = (r1 < n) ? x1[r1 * d + d0 + c] : 0.0f;
      lds2[ty][c] = (r2 < m) ? x2[r2 * d + d0 + c] : 0.0f;
    }
    if (ty == 0) {
      for (int c = tx; c < dc; c += TILE) ldsl[c] = inv_ls[d0 + c];
    }
    __syncthreads();
#pragma unroll 4
    for (int c = 0; c < dc; ++c) {
      const float diff = (lds1[ty][c] - lds2[tx][c]) * ldsl[c];
      acc = fmaf(diff, diff, acc);
    }
    __syncthreads();
  }

  const int row = row0 + ty;
  const int col = col0 + tx;
  if (row < n && col < m) {
    const float k = amp2 * matern52_of_d2(acc);
    out[row * m + col] = k;
    if (sym && row != col && col0 >= row0) {
      out[col * m + row] = k;
    }
  }
}

extern "C" void launch_gram_matern52(const float* x1, const float* x2,
                                     const float* inv_ls, float* out,
                                     int n, int m, int d, float amp2,
                                     int sym, hipStream_t stream) {
  const int tiles_n = (n + TILE - 1) / TILE;
  const int tiles_m = (m + TILE - 1) / TILE;
  dim3 grid(tiles_n * tiles_m);
  dim3 block(TILE * TILE);
  hipLaunchKernelGGL(gram_matern52_kernel, grid, block, 0, stream, x1, x2,
                     inv_ls, out, n, m, d, amp2, sym);
}

// Batched-restart gram for the ARD fit: one launch builds K for ALL
// restarts' lengthscale/amplitude sets, with the observation noise
// folded onto the diagonal and (optionally) the gradient factor
// G = amp^2 (5/3)(1 + sqrt5 r) e^{-sqrt5 r} emitted from the same
// distance pass. Replaces the torch-composed chain (scale -> GEMM
// d2 -> clamp -> sqrt -> exp -> mul -> add noise*I: seven (R, N, N)
// memory passes per NLL evaluation) with one fused kernel.
extern "C" __global__ __launch_bounds__(TILE * TILE) void
gram_matern52_batched_kernel(
    const float* __restrict__ x,        // (N, D)
    const float* __restrict__ inv_ls,   // (R, D)
    const float* __restrict__ amp2,     // (R,)
    const float* __restrict__ noise,    // (R,) diagonal addend
    float* __restrict__ kout,           // (R, N, N)
    float* __restrict__ gout,           // (R, N, N) or nullptr
    int r_count, int n, int d) {
  __shared__ float lds1[TILE][DCHUNK + 1];
  __shared__ float lds2[TILE][DCHUNK + 1];
  __shared__ float ldsl[DCHUNK];

  const int ty = threadIdx.x / TILE;
  const int tx = threadIdx.x % TILE;
  const int r = blockIdx.y;
  const float* ils = inv_ls + (long)r * d;
  float* K = kout + (long)r * n * n;
  float* G = (gout != nullptr) ? gout + (long)r * n * n : nullptr;

  int nwg = gridDim.x;
  int wg = blockIdx.x;
  if (nwg % 8 == 0) {
    const int cpx = nwg / 8;
    wg = (wg % 8) * cpx + wg / 8;
  }
  const int tiles_n = (n + TILE - 1) / TILE;
  const int row0 = (wg / tiles_n) * TILE;
  const int col0 = (wg % tiles_n) * TILE;
  if (row0 >= n || col0 >= n) return;
  if (col0 + TILE <= row0) return;  // symmetric: skip lower tiles

  float acc = 0.0f;
  for (int d0 = 0; d0 < d; d0 += DCHUNK) {
    const int dc = min(DCHUNK, d - d0);
    for (int c = tx; c < dc; c += TILE) {
      const int r1 = row0 + ty;
      const int r2 = col0 + ty;
      lds1[ty][c] = (r1 < n) ? x[r1 * d + d0 + c] : 0.0f;
      lds2[ty][c] = (r2 < n) ? x[r2 * d + d0 + c] : 0.0f;
    }
    if (ty == 0) {
      for (int c = tx; c < dc; c += TILE) ldsl[c] = ils[d0 + c];
    }
    __syncthreads();
#pragma unroll 4
    for (int c = 0; c < dc; ++c) {
      const float diff = (lds1[ty][c] - lds2[tx][c]) * ldsl[c];
      acc = fmaf(diff, diff, acc);
    }
    __syncthreads();
  }

  const int row = row0 + ty;
  const int col = col0 + tx;
  if (row < n && col < n) {
    const float a2 = amp2[r];
    const float sr = 2.2360679774997896f * sqrtf(fmaxf(acc, 0.0f));
    const float e = __expf(-sr);
    float k = a2 * (1.0f + sr + sr * sr * (1.0f / 3.0f)) * e;
    if (row == col) k += noise[r];
    K[(long)row * n + col] = k;
    if (row != col && col0 >= row0) K[(long)col * n + row] = k;
    if (G != nullptr) {
      const float g = a2 * (5.0f / 3.0f) * (1.0f + sr) * e;
      G[(long)row * n + col] = g;
      if (row != col && col0 >= row0) G[(long)col * n + row] = g;
    }
  }
}

extern "C" void launch_gram_matern52_batched(
    const float* x, const float* inv_ls, const float* amp2,
    const float* noise, float* kout, float* gout, int r, int n, int d,
    hipStream_t stream) {
  const int tiles_n = (n + TILE - 1) / TILE;
  dim3 grid(tiles_n * tiles_n, r);
  dim3 block(TILE * TILE);
  hipLaunchKernelGGL(gram_matern52_batched_kernel, grid, block, 0,
                     stream, x, inv_ls, amp2, noise, kout, gout, r, n,
                     d);
}
