// LDS-tiled fp8 (OCP e4m3) MFMA Matern-5/2 Gram kernel for gfx950.
//
// The 128x128 tile structure of gram_matern52_bf16_tiled.hip with the
// fp8 MFMA (__builtin_amdgcn_mfma_f32_16x16x32_fp8_fp8, i64-packed
// operands): one staging pass fills both 128x32 fp8 tiles (4 KiB each)
// via global_load_lds width-16; each wave does 8 x ds_read_b64 +
// 16 MFMAs per K-step. Host pre-scales inputs by 1/s (e4m3 range);
// the epilogue multiplies the dot back by s^2. gfx950 is OCP e4m3fn.

#include <hip/hip_runtime.h>
#include "common.h"

typedef __attribute__((ext_vector_type(4))) float f32x4;

#define BM 128
#define BN 128
#define BK 32
#define THREADS 256

extern "C" __global__ __launch_bounds__(THREADS) void
gram_matern52_fp8_tiled_kernel(
    const unsigned char* __restrict__ z1,  // (N, Dp) e4m3 bytes
    const unsigned char* __restrict__ z2,  // (M, Dp)
    const float* __restrict__ n1,          // (N,)
    const float* __restrict__ n2,          // (M,)
    float* __restrict__ out, int n, int m, int dp, float amp2,
    float scale2) {
  __shared__ unsigned char ldsA[BM * BK];  // 4 KiB, 32 B rows
  __shared__ unsigned char ldsB[BN * BK];

  const int tid = threadIdx.x;
  const int wave = tid / 64;
  const int lane = tid % 64;
  const int wr = wave >> 1;
  const int wc = wave & 1;

  const int tiles_m = (m + BN - 1) / BN;
  int wg = blockIdx.x;
  const int nwg = gridDim.x;
  if (nwg % 8 == 0) {
    const int cpx = nwg / 8;
    wg = (wg % 8) * cpx + wg / 8;
  }
  const int row0 = (wg / tiles_m) * BM;
  const int col0 = (wg % tiles_m) * BN;
  if (row0 >= n || col0 >= m) return;

  f32x4 acc[4][4];
#pragma unroll
  for (int i = 0; i < 4; ++i)
#pragma unroll
    for (int j = 0; j < 4; ++j)
      acc[i][j] = f32x4{0.f, 0.f, 0.f, 0.f};

  // One pass: wave `w` fills LDS bytes [w*1024, +1024); lane l supplies
  // bytes [l*16, +16) => row = w*32 + l/2, k-seg = (l%2)*16.
  const int a_row = min(row0 + wave * 32 + lane / 2, n - 1);
  const int b_row = min(col0 + wave * 32 + lane / 2, m - 1);
  const int kseg = (lane % 2) * 16;

  for (int k0 = 0; k0 < dp; k0 += BK) {
    __builtin_amdgcn_global_load_lds(
        (const __attribute__((address_space(1))) unsigned int*)
            (z1 + (long)a_row * dp + k0 + kseg),
        (__attribute__((address_space(3))) unsigned int*)
            (ldsA + wave * 1024 + lane * 16),
        16, 0, 0);
    __builtin_amdgcn_global_load_lds(
        (const __attribute__((address_space(1))) unsigned int*)
            (z2 + (long)b_row * dp + k0 + kseg),
        (__attribute__((address_space(3))) unsigned int*)
            (ldsB + wave * 1024 + lane * 16),
        16, 0, 0);
    __builtin_amdgcn_s_waitcnt(0);
    __syncthreads();

    long aF[4], bF[4];
#pragma unroll
    for (int f = 0; f < 4; ++f) {
      const int arow = wr * 64 + f * 16 + (lane & 15);
      const int brow = wc * 64 + f * 16 + (lane & 15);
      const int kb = (lane >> 4) * 8;
      aF[f] = *reinterpret_cast<const long*>(ldsA + arow * BK + kb);
      bF[f] = *reinterpret_cast<const long*>(ldsB + brow * BK + kb);
    }
#pragma unroll
    for (int i = 0; i < 4; ++i)
#pragma unroll
      for (int j = 0; j < 4; ++j)
        acc[i][j] = __builtin_amdgcn_mfma_f32_16x16x32_fp8_fp8(
            aF[i], bF[j], acc[i][j], 0, 0, 0);
    __syncthreads();
  }

#pragma unroll
  for (int i = 0; i < 4; ++i) {
#pragma unroll
    for (int j = 0; j < 4; ++j) {
      const int col = col0 + wc * 64 + j * 16 + (lane & 15);
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const int row = row0 + wr * 64 + i * 16 + (lane >> 4) * 4 + r;
        if (row < n && col < m) {
          const float d2 = fmaxf(
              n1[row] + n2[col] - 2.0f * scale2 * acc[i][j][r], 0.0f);
          out[(long)row * m + col] = amp2 * matern52_of_d2(d2);
        }
      }
    }
  }
}

extern "C" void launch_gram_matern52_fp8_tiled(
    const unsigned char* z1, const unsigned char* z2, const float* n1,
    const float* n2, float* out, int n, int m, int dp, float amp2,
    float scale2, hipStream_t stream) {
  const int tiles_n = (n + BM - 1) / BM;
  const int tiles_m = (m + BN - 1) / BN;
  hipLaunchKernelGGL(gram_matern52_fp8_tiled_kernel,
                     dim3(tiles_n * tiles_m), dim3(THREADS), 0, stream,
                     z1, z2, n1, n2, out, n, m, dp, amp2, scale2);
}
