"""HIP batched Cholesky vs MAGMA/torch at ladder shapes."""
import sys, time
sys.path.insert(0, '.')
import torch
from vizier_amd._src.ops import dispatch
ext = dispatch.require_ext()

def bench(fn, iters=20):
  fn(); torch.cuda.synchronize()
  t0 = time.perf_counter()
  for _ in range(iters): fn()
  torch.cuda.synchronize()
  return (time.perf_counter() - t0) / iters * 1e3

for (b, n) in [(12, 1000), (16, 1000), (4, 1000), (12, 500), (25, 1241)]:
  g = torch.Generator().manual_seed(0)
  a = torch.randn(b, n, 64, generator=g)
  K = (a @ a.mT + n * torch.eye(n)).cuda()
  t_hip = bench(lambda: ext.batched_cholesky(K.clone()))
  t_torch = bench(lambda: torch.linalg.cholesky_ex(K))
  print(f'B={b} N={n}: HIP {t_hip:.2f} ms | torch/MAGMA {t_torch:.2f} ms '
        f'| speedup {t_torch/t_hip:.1f}x', flush=True)
