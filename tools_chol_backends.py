"""Times batched fp32 Cholesky backends at the fit shapes.

Compares: custom v2 (multi-launch), custom v3 (cooperative, opt-in via
env), torch-MAGMA, torch-hipSOLVER — at (R=3, N=1000) grad-eval and
(R=12, N=1000) line-search-ladder shapes. Each backend in-process
except the preferred_linalg_library switch (process-global, so run
MAGMA first, then hipSOLVER via a subprocess if needed).
"""
import sys
import time

import torch

sys.path.insert(0, '.')
from vizier_amd._src.ops import dispatch as ops  # noqa: E402


def bench(fn, n_iter=20):
  fn()
  torch.cuda.synchronize()
  t0 = time.perf_counter()
  for _ in range(n_iter):
    fn()
  torch.cuda.synchronize()
  return (time.perf_counter() - t0) / n_iter * 1e3


def make_k(r, n):
  g = torch.Generator().manual_seed(0)
  a = torch.randn(r, n, 48, generator=g)
  return (a @ a.mT / 48 + torch.eye(n)).cuda().contiguous()


ext = ops.require_ext()
which = sys.argv[1] if len(sys.argv) > 1 else 'all'
for r, n in ((3, 1000), (12, 1000), (3, 2000)):
  k = make_k(r, n)
  if which in ('all', 'custom'):
    ms = bench(lambda: ext.batched_potrf(k))
    print(f'R={r:3d} N={n}: custom_v2      {ms:8.3f} ms', flush=True)
  if which in ('all', 'magma'):
    torch.backends.cuda.preferred_linalg_library('magma')
    ms = bench(lambda: torch.linalg.cholesky_ex(k))
    print(f'R={r:3d} N={n}: torch-magma    {ms:8.3f} ms', flush=True)
  if which == 'hipsolver':
    torch.backends.cuda.preferred_linalg_library('cusolver')
    ms = bench(lambda: torch.linalg.cholesky_ex(k))
    print(f'R={r:3d} N={n}: torch-hipsolver{ms:8.3f} ms', flush=True)
  if which == 'empty':
    # Launch-floor probe: 63 dependent tiny torch kernels (same count
    # as v2's per-potrf launches) on a 1-element tensor.
    x = torch.zeros(1, device='cuda')

    def chain():
      for _ in range(63):
        x.add_(1.0)
    ms = bench(chain)
    print(f'R={r:3d} N={n}: 63-launch chain{ms:8.3f} ms '
          f'({ms / 63 * 1e3:.1f} us/launch)', flush=True)
  if which == 'trsv':
    L = torch.linalg.cholesky(k).contiguous()
    b = torch.randn(r, n).cuda().contiguous()
    ms = bench(lambda: ext.batched_trsv_lower(L, b))
    print(f'R={r:3d} N={n}: custom_trsv    {ms:8.3f} ms', flush=True)
    ms = bench(lambda: torch.linalg.solve_triangular(
        L, b.unsqueeze(-1), upper=False))
    print(f'R={r:3d} N={n}: rocblas_trsv   {ms:8.3f} ms', flush=True)
    eye = torch.eye(n, device='cuda')
    ms = bench(lambda: torch.linalg.solve_triangular(
        L, eye.expand(r, n, n), upper=False))
    print(f'R={r:3d} N={n}: trsm_inv(NxN)  {ms:8.3f} ms', flush=True)
