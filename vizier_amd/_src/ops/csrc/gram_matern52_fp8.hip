// fp8 (OCP e4m3) MFMA Matern-5/2 Gram kernel for gfx950 (config 5).
//
// Same tile structure as gram_matern52_bf16.hip but the cross-term GEMM
// uses __builtin_amdgcn_mfma_f32_16x16x32_fp8_fp8 (8 fp8 per lane,
// packed as i64). Inputs are pre-scaled by 1/s on the host so |z| stays
// well inside e4m3 range; the kernel multiplies the dot product back by
// s^2. Norms come from the fp8-rounded values (exact rounded distance).
//
// NOTE: gfx950 uses OCP e4m3fn (not MI300X fnuz) — torch's
// float8_e4m3fn matches (guide §4).

#include <hip/hip_runtime.h>
#include "common.h"

typedef __attribute__((ext_vector_type(4))) float f32x4;

#define WAVES_PER_BLOCK 4

extern "C" __global__ __launch_bounds__(WAVES_PER_BLOCK * 64) void
gram_matern52_fp8_kernel(const unsigned char* __restrict__ z1,  // (N, Dp)
                         const unsigned char* __restrict__ z2,  // (M, Dp)
                         const float* __restrict__ n1,          // (N,)
                         const float* __restrict__ n2,          // (M,)
                         float* __restrict__ out, int n, int m, int dp,
                         float amp2, float scale2 /* s^2 */) {
  const int wave = threadIdx.x / 64;
  const int lane = threadIdx.x % 64;
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
    const long a = *reinterpret_cast<const long*>(
        z1 + (long)a_row * dp + k0 + k_base);
    const long b = *reinterpret_cast<const long*>(
        z2 + (long)b_col * dp + k0 + k_base);
    acc = __builtin_amdgcn_mfma_f32_16x16x32_fp8_fp8(a, b, acc, 0, 0, 0);
  }

#pragma unroll
  for (int r = 0; r < 4; ++r) {
    const int row = row0 + (lane >> 4) * 4 + r;
    const int col = col0 + (lane & 15);
    if (row < n && col < m) {
      const float d2 = fmaxf(n1[row] + n2[col] - 2.0f * scale2 * acc[r],
                             0.0f);
      out[(long)row * m + col] = amp2 * matern52_of_d2(d2);
    }
  }
}

extern "C" void launch_gram_matern52_fp8(
    const unsigned char* z1, const unsigned char* z2, const float* n1,
    const float* n2, float* out, int n, int m, int dp, float amp2,
    float scale2, hipStream_t stream) {
  const int tiles_n = (n + 15) / 16;
  const int tiles_m = (m + 16 * WAVES_PER_BLOCK - 1) /
                      (16 * WAVES_PER_BLOCK);
  hipLaunchKernelGGL(gram_matern52_fp8_kernel, dim3(tiles_n * tiles_m),
                     dim3(WAVES_PER_BLOCK * 64), 0, stream, z1, z2, n1,
                     n2, out, n, m, dp, amp2, scale2);
}
