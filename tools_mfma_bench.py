"""Large-N Gram kernel throughput: fp32 vector vs bf16/fp8 MFMA paths.

Measures the big-sweep regime (config 4/5 scale: N x M cross-Grams with
hundreds of features). Cross-term FLOPs = 2*N*M*D; run under
rocprofv3 --pmc MfmaUtil to verify the matrix cores are engaged.
"""

import sys
import time

import torch

sys.path.insert(0, '.')
from vizier_amd._src.ops import dispatch as ops

ext = ops.require_ext()


def bench(fn, iters=10):
  for _ in range(3):
    fn()
  torch.cuda.synchronize()
  t0 = time.perf_counter()
  for _ in range(iters):
    fn()
  torch.cuda.synchronize()
  return (time.perf_counter() - t0) / iters


def main():
  for (n, m, d) in [(4096, 4096, 256), (8192, 8192, 512),
                    (16384, 8192, 128)]:
    g = torch.Generator().manual_seed(0)
    x1 = torch.rand(n, d, generator=g).cuda()
    x2 = torch.rand(m, d, generator=g).cuda()
    ls = (torch.rand(d, generator=g) + 0.5).cuda()
    flops = 2.0 * n * m * d
    t32 = bench(lambda: ext.gram_matern52(x1, x2, ls, 1.0))
    t16 = bench(lambda: ext.gram_matern52_bf16(x1, x2, ls, 1.0))
    tt = bench(lambda: ext.gram_matern52_bf16_tiled(x1, x2, ls, 1.0))
    t8 = bench(lambda: ext.gram_matern52_fp8(x1, x2, ls, 1.0))
    t8t = bench(lambda: ext.gram_matern52_fp8_tiled(x1, x2, ls, 1.0))
    k32 = ext.gram_matern52(x1, x2, ls, 1.0)
    k16 = ext.gram_matern52_bf16(x1, x2, ls, 1.0)
    kt = ext.gram_matern52_bf16_tiled(x1, x2, ls, 1.0)
    k8 = ext.gram_matern52_fp8(x1, x2, ls, 1.0)
    print(f'N={n} M={m} D={d}: '
          f'fp32 {t32*1e3:.2f}ms ({flops/t32/1e12:.1f} TF) | '
          f'bf16 MFMA {t16*1e3:.2f}ms ({flops/t16/1e12:.1f} TF) | '
          f'bf16 LDS-tiled {tt*1e3:.2f}ms ({flops/tt/1e12:.1f} TF) | '
          f'fp8 MFMA {t8*1e3:.2f}ms ({flops/t8/1e12:.1f} TF) | '
          f'fp8 LDS-tiled {t8t*1e3:.2f}ms ({flops/t8t/1e12:.1f} TF) | '
          f'bf16 err {float((k16-k32).abs().max()):.4f} '
          f'tiled-vs-strip err {float((kt-k16).abs().max()):.2e} '
          f'fp8 err {float((k8-k32).abs().max()):.4f}', flush=True)


if __name__ == '__main__':
  main()
