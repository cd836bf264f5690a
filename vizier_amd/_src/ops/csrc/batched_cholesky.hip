// Batched fp32 Cholesky (lower) for gfx950 — the ARD line-search path.
//
// MAGMA's batched POTRF spends most of its time in unblocked spotf2
// panel kernels plus per-call pointer-displacement launches
// (profiles/bench_kernel_stats2.csv: ~16% of a suggest). The ARD
// line-search ladder only needs LOSS VALUES (no autograd), so this
// hand-written forward replaces it there: one 256-thread workgroup per
// batch matrix, right-looking blocked algorithm with the CURRENT panel
// staged in dynamic LDS (row stride padded to 33 floats — a stride of
// 32 would put every panel row in the same LDS bank):
//   per 32-wide panel at column j0:
//     1. factor the 32x32 diagonal block (block-wide: scale column,
//        rank-1 update, 3 barriers per column),
//     2. L21 = A21 * L11^-T: each thread forward-substitutes one row
//        in LDS,
//     3. trailing update A22 -= L21 L21^T: threads own 2x2 output
//        tiles, reading L21 rows from LDS.
// The gradient path keeps torch's cholesky (autograd); N is capped by
// the 160 KiB LDS (the binding falls back to torch beyond it).
//
// Writes the lower triangle in place (upper cleared); info[b] = 0 on
// success or the 1-based column of the first non-positive pivot.

#include <hip/hip_runtime.h>

#include "common.h"

#define NB 32
#define PST 33  // padded LDS row stride (floats)
#define CHOL_BLOCK 256

extern "C" __global__ __launch_bounds__(CHOL_BLOCK) void
batched_cholesky_kernel(float* __restrict__ a,  // (B, N, N) in place
                        int* __restrict__ info, int n, long mat_stride) {
  extern __shared__ float panel[];  // (n rows) x PST floats
  __shared__ int bad_s;
  const int tid = threadIdx.x;
  float* A = a + (long)blockIdx.x * mat_stride;
  if (tid == 0) bad_s = 0;
  __syncthreads();

  for (int j0 = 0; j0 < n; j0 += NB) {
    const int nb = min(NB, n - j0);
    const int rows = n - j0;
    // Stage panel columns [j0, j0+nb), rows [j0, n) into LDS.
    for (int e = tid; e < rows * nb; e += CHOL_BLOCK) {
      const int r = e / nb, c = e % nb;
      panel[r * PST + c] = A[(long)(j0 + r) * n + j0 + c];
    }
    __syncthreads();

    // 1) Factor the nb x nb diagonal block (all threads participate).
    for (int j = 0; j < nb; ++j) {
      if (tid == 0) {
        float d = panel[j * PST + j];
        if (d <= 0.0f || !isfinite(d)) {
          if (bad_s == 0) bad_s = j0 + j + 1;
          d = 1.0f;  // keep going; caller discards via info
        }
        panel[j * PST + j] = sqrtf(d);
      }
      __syncthreads();
      const float inv = 1.0f / panel[j * PST + j];
      for (int r = j + 1 + tid; r < nb; r += CHOL_BLOCK) {
        panel[r * PST + j] *= inv;
      }
      __syncthreads();
      const int rem = nb - j - 1;
      for (int e = tid; e < rem * rem; e += CHOL_BLOCK) {
        const int r = j + 1 + e / rem;
        const int c = j + 1 + e % rem;
        if (c <= r) {
          panel[r * PST + c] -= panel[r * PST + j] * panel[c * PST + j];
        }
      }
      __syncthreads();
    }

    // 2) L21 = A21 * L11^-T: thread r substitutes its row in place.
    for (int r = nb + tid; r < rows; r += CHOL_BLOCK) {
      for (int c = 0; c < nb; ++c) {
        float v = panel[r * PST + c];
        for (int kk = 0; kk < c; ++kk) {
          v -= panel[r * PST + kk] * panel[c * PST + kk];
        }
        panel[r * PST + c] = v / panel[c * PST + c];
      }
    }
    __syncthreads();

    // Write the factored panel back (upper of the diagonal block = 0).
    for (int e = tid; e < rows * nb; e += CHOL_BLOCK) {
      const int r = e / nb, c = e % nb;
      float v = panel[r * PST + c];
      if (r < nb && c > r) v = 0.0f;
      A[(long)(j0 + r) * n + j0 + c] = v;
    }
    __syncthreads();

    // 3) Trailing update: A22 -= L21 L21^T (lower triangle, 2x2 tiles).
    const int trows = rows - nb;
    if (trows > 0) {
      const int t2 = (trows + 1) / 2;
      for (int e = tid; e < t2 * t2; e += CHOL_BLOCK) {
        const int ri = (e / t2) * 2, ci = (e % t2) * 2;
        if (ci > ri + 1) continue;  // strictly-upper tile
        const bool has_r1 = (ri + 1 < trows);
        const bool has_c1 = (ci + 1 < trows);
        float acc00 = 0.f, acc01 = 0.f, acc10 = 0.f, acc11 = 0.f;
        const float* lr0 = panel + (nb + ri) * PST;
        const float* lr1 = panel + (nb + (has_r1 ? ri + 1 : ri)) * PST;
        const float* lc0 = panel + (nb + ci) * PST;
        const float* lc1 = panel + (nb + (has_c1 ? ci + 1 : ci)) * PST;
#pragma unroll 8
        for (int kk = 0; kk < nb; ++kk) {
          const float r0 = lr0[kk], r1 = lr1[kk];
          const float c0 = lc0[kk], c1 = lc1[kk];
          acc00 = fmaf(r0, c0, acc00);
          acc01 = fmaf(r0, c1, acc01);
          acc10 = fmaf(r1, c0, acc10);
          acc11 = fmaf(r1, c1, acc11);
        }
        const int gr0 = j0 + nb + ri, gc0 = j0 + nb + ci;
        if (gc0 <= gr0) A[(long)gr0 * n + gc0] -= acc00;
        if (has_c1 && gc0 + 1 <= gr0) A[(long)gr0 * n + gc0 + 1] -= acc01;
        if (has_r1) {
          if (gc0 <= gr0 + 1) A[(long)(gr0 + 1) * n + gc0] -= acc10;
          if (has_c1 && gc0 + 1 <= gr0 + 1) {
            A[(long)(gr0 + 1) * n + gc0 + 1] -= acc11;
          }
        }
      }
    }
    __syncthreads();
  }

  if (tid == 0) info[blockIdx.x] = bad_s;
}

// Max N such that the n x PST fp32 panel fits in 160 KiB of LDS.
extern "C" int batched_cholesky_max_n(void) {
  return (160 * 1024) / (PST * (int)sizeof(float));  // 1241
}

extern "C" int launch_batched_cholesky(float* a, int* info, int batch,
                                       int n, hipStream_t stream) {
  const size_t shmem = (size_t)n * PST * sizeof(float);
  static int attr_set = 0;
  if (!attr_set) {
    // May be unsupported for this symbol form on some ROCm builds;
    // harmless either way, but it leaves a sticky error code that a
    // later hipGetLastError would misattribute to the launch — clear.
    (void)hipFuncSetAttribute(
        (const void*)batched_cholesky_kernel,
        hipFuncAttributeMaxDynamicSharedMemorySize, 160 * 1024);
    (void)hipGetLastError();
    attr_set = 1;
  }
  hipLaunchKernelGGL(batched_cholesky_kernel, dim3(batch),
                     dim3(CHOL_BLOCK), shmem, stream, a, info, n,
                     (long)n * n);
  return (int)hipGetLastError();
}
