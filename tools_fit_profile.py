"""Breakdown of the ARD fit cost at bench shape (N=1000, D=20)."""
import sys, time
import torch
sys.path.insert(0, '.')
from vizier_amd._src.gp import gp_model
from vizier_amd._src.gp.matern import gram_matern52

dev = 'cuda'
g = torch.Generator().manual_seed(0)
x = torch.rand(1000, 20, generator=g).to(dev)
y = torch.sin(3 * x[:, 0]).to(dev)

def t(fn, iters=20):
    for _ in range(3): fn()
    torch.cuda.synchronize(); t0 = time.perf_counter()
    for _ in range(iters): fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters * 1e3

for R in (5, 25):
    raw = torch.randn(R, 23, device=dev) * 0.5
    params = gp_model.GPParams.from_raw(raw)
    K = gram_matern52(x.unsqueeze(0), None, params.lengthscales,
                      params.amplitude)
    K = K + params.noise.reshape(-1,1,1) * torch.eye(1000, device=dev)
    print(f'R={R}: gram={t(lambda: gram_matern52(x.unsqueeze(0), None, params.lengthscales, params.amplitude)):.2f}ms '
          f'cholesky={t(lambda: torch.linalg.cholesky_ex(K)):.2f}ms', flush=True)
    L, _ = torch.linalg.cholesky_ex(K)
    b = torch.rand(R, 1000, 1, device=dev)
    print(f'   trsm x2={t(lambda: gp_model._chol_solve(L, b)):.2f}ms', flush=True)

def nll_fwd():
    raw = torch.randn(25, 23, device=dev) * 0.5
    with torch.no_grad():
        gp_model.negative_log_marginal_likelihood(raw, x, y)
def nll_bwd():
    raw = (torch.randn(5, 23, device=dev) * 0.5).requires_grad_(True)
    loss = gp_model.negative_log_marginal_likelihood(raw, x, y)
    torch.autograd.grad(loss.sum(), raw)
print(f'NLL fwd (R=25): {t(nll_fwd):.2f}ms', flush=True)
print(f'NLL fwd+bwd (R=5): {t(nll_bwd):.2f}ms', flush=True)

def full_fit():
    gp_model.train_gp(x, y, num_restarts=4, max_iters=20, seed=1)
print(f'train_gp warm-equivalent (20 iters): {t(full_fit, iters=3):.1f}ms', flush=True)
