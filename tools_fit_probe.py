"""Isolate the fit-time regression: hipSOLVER pin vs fp64 cache build."""
import sys, time
sys.path.insert(0, '.')
import torch

def bench(fn, iters=5):
  fn(); torch.cuda.synchronize()
  t0 = time.perf_counter()
  for _ in range(iters): fn()
  torch.cuda.synchronize()
  return (time.perf_counter() - t0) / iters

def main():
  from vizier_amd._src.gp import gp_model  # pins cusolver at import
  g = torch.Generator().manual_seed(0)
  x = torch.rand(1000, 20, generator=g).cuda()
  y = torch.randn(1000, generator=g).cuda()
  raw = torch.randn(6, 23, generator=g).cuda() * 0.5

  def nll_fwdbwd():
    r = raw.detach().requires_grad_(True)
    loss = gp_model.negative_log_marginal_likelihood(r, x, y)
    torch.autograd.grad(loss.sum(), r)

  for backend in ('cusolver', 'magma', 'default'):
    torch.backends.cuda.preferred_linalg_library(backend)
    try:
      t = bench(nll_fwdbwd)
      print(f'{backend}: NLL fwd+bwd (R=6,N=1000) {t*1e3:.1f} ms', flush=True)
    except Exception as e:
      print(f'{backend}: FAILED {e!r}', flush=True)

  torch.backends.cuda.preferred_linalg_library('cusolver')
  from vizier_amd._src.gp.matern import gram_matern52
  params = gp_model.GPParams.from_raw(raw[0])

  def cache_build(dtype):
    x_ = x.to(dtype)
    ls = params.lengthscales.detach().to(dtype)
    amp = params.amplitude.detach().to(dtype)
    K = gram_matern52(x_, None, ls, amp)
    K = K + 1e-3 * torch.eye(1000, dtype=dtype, device='cuda')
    L = torch.linalg.cholesky_ex(K)[0]
    eye = torch.eye(1000, dtype=dtype, device='cuda')
    z = torch.linalg.solve_triangular(L, eye, upper=False)
    (z.T @ z).float()

  for dtype in (torch.float32, torch.float64):
    t = bench(lambda: cache_build(dtype))
    print(f'cache build {dtype}: {t*1e3:.1f} ms', flush=True)

  # full train_gp timings
  for backend in ('cusolver', 'default'):
    torch.backends.cuda.preferred_linalg_library(backend)
    t = bench(lambda: gp_model.train_gp(x, y, num_restarts=2,
                                        max_iters=12, seed=1,
                                        warm_start_raw=raw[0]), iters=3)
    print(f'train_gp warm ({backend}): {t*1e3:.1f} ms', flush=True)

if __name__ == '__main__':
  main()
