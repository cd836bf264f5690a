// bf16 MFMA Matern-5/2 Gram kernel for gfx950 (matrix cores).
//
// Computes K[i,j] = amp^2 * m52(sqrt(n1[i] + n2[j] - 2*z1_i . z2_j))
// where z = x / lengthscales. The cross-term GEMM z1 @ z2^T runs on the
// CDNA4 matrix cores via __builtin_amdgcn_mfma_f32_16x16x32_bf16 (one
// 16x16 output tile per 64-lane wave, K in steps of 32); the row norms
// n1/n2 are precomputed in fp32 on the host side of the binding, so the
// bf16 rounding only affects the cross term. Inputs are bf16 (ushort
// bit pattern), zero-padded to a multiple of 32 columns.
//
// Fragment layouts (guide §3, HW-verified in learn_hip m89/m91):
//   A (16x32): lane l holds 8 contiguous bf16 at row = l&15,
//              k = (l>>4)*8 .. +7.
//   B (32x16): lane l holds 8 contiguous bf16 at col = l&15 (same
//              pattern as A because both operands index the non-K dim
//              by l&15 — we feed z2 row-major, giving z1 @ z2^T).
//   C/D (16x16 fp32x4): col = l&15, row = (l>>4)*4 + reg.

#include <hip/hip_runtime.h>
#include "common.h"

typedef __attribute__((ext_vector_type(8))) short bf16x8;
typedef __attribute__((ext_vector_type(4))) float f32x4;

#define WAVES_PER_BLOCK 4

extern "C" __global__ __launch_bounds__(WAVES_PER_BLOCK * 64) void
gram_matern52_bf16_kernel(const unsigned short* __restrict__ z1,  // (N, Dp)
                          const unsigned short* __restrict__ z2,  // (M, Dp)
                          const float* __restrict__ n1,           // (N,)
                          const float* __restrict__ n2,           // (M,)
                          float* __restrict__ out, int n, int m, int dp,
                          float amp2) {
  const int wave = threadIdx.x / 64;
  const int lane = threadIdx.x % 64;
  // Each block covers a 16 x (16*WAVES) strip; XCD-aware swizzle (T1).
  const int tiles_m = (m + 16 * WAVES_PER_BLOCK - 1) /
                      (16 * WAVES_PER_BLOCK);
  int wg = blockIdx.x;
  const int nwg = gridDim.x;
  if (nwg % 8 == 0) {
    const int cpx = nwg / 8;
    wg = (wg % 8) * cpx + wg / 8;
  }
  const int row0 = (wg / tiles_m) * 16;
  const int col0 = (wg % tiles_m) * (16 * WAVES_PER_BLOCK) + wave * 16;
  if (row0 >= n || col0 >= m) return;

  const int a_row = min(row0 + (lane & 15), n - 1);
  const int b_col = min(col0 + (lane & 15), m - 1);
  const int k_base = (lane >> 4) * 8;

  f32x4 acc = {0.f, 0.f, 0.f, 0.f};
  for (int k0 = 0; k0 < dp; k0 += 32) {
    bf16x8 a = *reinterpret_cast<const bf16x8*>(
        z1 + (long)a_row * dp + k0 + k_base);
    bf16x8 b = *reinterpret_cast<const bf16x8*>(
        z2 + (long)b_col * dp + k0 + k_base);
    acc = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b, acc, 0, 0, 0);
  }

#pragma unroll
  for (int r = 0; r < 4; ++r) {
    const int row = row0 + (lane >> 4) * 4 + r;
    const int col = col0 + (lane & 15);
    if (row < n && col < m) {
      const float d2 = fmaxf(n1[row] + n2[col] - 2.0f * acc[r], 0.0f);
      out[(long)row * m + col] = amp2 * matern52_of_d2(d2);
    }
  }
}

extern "C" void launch_gram_matern52_bf16(
    const unsigned short* z1, const unsigned short* z2, const float* n1,
    const float* n2, float* out, int n, int m, int dp, float amp2,
    hipStream_t stream) {
  const int tiles_n = (n + 15) / 16;
  const int tiles_m = (m + 16 * WAVES_PER_BLOCK - 1) /
                      (16 * WAVES_PER_BLOCK);
  hipLaunchKernelGGL(gram_matern52_bf16_kernel, dim3(tiles_n * tiles_m),
                     dim3(WAVES_PER_BLOCK * 64), 0, stream, z1, z2, n1,
                     n2, out, n, m, dp, amp2);
}
