"""Matérn-5/2 + linear kernel GP (the reference's `linear_coef` option).

Parity with tuned_gp_models.py:204-234: when `linear_coef` is set, the
ARD Matérn-5/2 kernel gains an additive continuous-only linear term

  k_lin(x, y) = (lc * s)^2 * sum_d (x_d/l_d - lc*c)(y_d/l_d - lc*c)

with a log-uniform-initialized slope amplitude `s` (SoftClip-bounded
like the Matérn amplitude, regularized 0.01*log(s/0.039)^2) and a
normal-initialized scalar shift `c` (regularized 0.5*c^2), both fitted
jointly with the usual hyperparameters by batched L-BFGS. The linear
term captures global trends that a stationary kernel models poorly
(high-D or drifting objectives).

MI355X note: posteriors from this model deliberately publish
K_inv=None so ScoringFunction routes through predict() (the combined
kernel) instead of the fused pure-Matérn HIP scorers.
"""

from __future__ import annotations

import dataclasses
import math
from typing import Optional, Tuple

import torch

from vizier_amd._src.gp import lbfgs
from vizier_amd._src.gp.gp_model import (
    _LOG_AMP_BOUNDS,
    _LOG_LS_BOUNDS,
    _LOG_NOISE_BOUNDS,
    _MEAN_BOUNDS,
    _from_bounded,
    _to_bounded,
    cholesky_with_jitter,
    safe_cholesky_ex,
)
from vizier_amd._src.gp.matern import gram_matern52

# Raw layout per restart: [amp, noise, mean, slope, shift, ls_0..ls_D-1]
_N_EXTRA = 5  # non-lengthscale columns


@dataclasses.dataclass
class LinearMaternParams:
  amplitude: torch.Tensor     # (...,)
  noise: torch.Tensor         # (...,)
  mean: torch.Tensor          # (...,)
  slope: torch.Tensor         # (...,) linear slope amplitude s
  shift: torch.Tensor         # (...,) linear shift c (unconstrained)
  lengthscales: torch.Tensor  # (..., D)

  @classmethod
  def from_raw(cls, raw: torch.Tensor) -> 'LinearMaternParams':
    return cls(
        amplitude=_to_bounded(raw[..., 0], *_LOG_AMP_BOUNDS).exp(),
        noise=_to_bounded(raw[..., 1], *_LOG_NOISE_BOUNDS).exp(),
        mean=_to_bounded(raw[..., 2], *_MEAN_BOUNDS),
        slope=_to_bounded(raw[..., 3], *_LOG_AMP_BOUNDS).exp(),
        shift=raw[..., 4],
        lengthscales=_to_bounded(raw[..., 5:], *_LOG_LS_BOUNDS).exp())


def _combined_gram(params: LinearMaternParams, linear_coef: float,
                   x1: torch.Tensor, x2: Optional[torch.Tensor]
                   ) -> torch.Tensor:
  """Matérn-5/2 + scaled linear kernel; x1 (..., N, D), x2 (..., M, D)."""
  K = gram_matern52(x1, x2, params.lengthscales, params.amplitude)
  ls = params.lengthscales.unsqueeze(-2)              # (..., 1, D)
  shift = (linear_coef * params.shift)[..., None, None]
  z1 = x1 / ls - shift
  z2 = z1 if x2 is None else x2 / ls - shift
  s2 = (linear_coef * params.slope).square()[..., None, None]
  return K + s2 * (z1 @ z2.mT)


def negative_log_marginal_likelihood(
    raw: torch.Tensor, x: torch.Tensor, y: torch.Tensor,
    linear_coef: float) -> torch.Tensor:
  """Batched NLL + the reference's slope/shift regularizers."""
  params = LinearMaternParams.from_raw(raw)
  n = x.shape[0]
  K = _combined_gram(params, linear_coef, x.unsqueeze(0), None)
  noise = params.noise.reshape(-1, 1, 1)
  K = K + noise * torch.eye(n, dtype=x.dtype, device=x.device)
  L, info = safe_cholesky_ex(K)
  resid = (y.unsqueeze(0) - params.mean.unsqueeze(-1)).unsqueeze(-1)
  z = torch.linalg.solve_triangular(L, resid, upper=False)
  quad = (z * z).sum(dim=(-1, -2))
  logdet = 2.0 * torch.log(torch.diagonal(L, dim1=-2, dim2=-1)).sum(-1)
  nll = 0.5 * (quad + logdet + n * math.log(2 * math.pi))
  # Priors: tuned_gp_models.py:212 (slope), :217 (shift); the other
  # columns keep the main GP's mild raw-space pull.
  nll = nll + 0.01 * torch.log(params.slope / 0.039).square()
  nll = nll + 0.5 * params.shift.square()
  other = torch.cat([raw[:, :4], raw[:, 5:]], dim=1)
  nll = nll + 0.01 * (other * other).sum(-1)
  bad = info != 0
  return torch.where(bad, torch.full_like(nll, float('inf')), nll)


@dataclasses.dataclass
class LinearMaternPosterior:
  """Cached posterior for the combined kernel.

  Duck-type compatible with gp_model.GPPosterior where the designers
  need it; K_inv is always None (see module pydoc)."""

  x: torch.Tensor
  params: LinearMaternParams
  linear_coef: float
  L: torch.Tensor
  alpha: torch.Tensor        # (N,)
  nll: float
  raw: Optional[torch.Tensor] = None
  aux: Optional[torch.Tensor] = None  # [slope, shift] for broadcasts

  @property
  def K_inv(self):
    return None

  def predict(self, xq: torch.Tensor) -> Tuple[torch.Tensor, torch.Tensor]:
    k = _combined_gram(self.params, self.linear_coef, xq, self.x)
    mean = self.params.mean + k @ self.alpha
    ls = self.params.lengthscales
    zq = xq / ls - self.linear_coef * self.shift_value
    prior = self.params.amplitude.square() + \
        (self.linear_coef * self.params.slope).square() * \
        (zq * zq).sum(-1)
    v = torch.linalg.solve_triangular(self.L, k.T, upper=False)
    var = prior - (v * v).sum(0)
    return mean, var.clamp_min(1e-12).sqrt()

  @property
  def shift_value(self) -> torch.Tensor:
    return self.params.shift


def train_linear_matern_gp(
    x: torch.Tensor, y: torch.Tensor, *, linear_coef: float,
    num_restarts: int = 4, max_iters: int = 50, seed: int = 0,
    warm_start_raw: Optional[torch.Tensor] = None
    ) -> LinearMaternPosterior:
  """Fits the combined-kernel GP (restarted batched L-BFGS)."""
  x = x.detach()
  y = y.detach().reshape(-1)
  n, d = x.shape
  g = torch.Generator(device='cpu').manual_seed(seed)
  u = torch.rand(num_restarts + 1, d + _N_EXTRA, generator=g) * 0.9 + .05
  raw0 = torch.log(u / (1 - u))
  raw0[:, 4] = torch.randn(num_restarts + 1, generator=g)  # shift ~ N(0,1)
  raw0[0, 0] = _from_bounded(math.log(1.0), *_LOG_AMP_BOUNDS)
  raw0[0, 1] = _from_bounded(math.log(1e-4), *_LOG_NOISE_BOUNDS)
  raw0[0, 2] = _from_bounded(0.0, *_MEAN_BOUNDS)
  raw0[0, 3] = _from_bounded(math.log(0.039), *_LOG_AMP_BOUNDS)
  raw0[0, 4] = 0.0
  raw0[0, 5:] = _from_bounded(math.log(0.5), *_LOG_LS_BOUNDS)
  raw0 = raw0.to(device=x.device, dtype=x.dtype)
  if warm_start_raw is not None and \
      warm_start_raw.numel() == d + _N_EXTRA:
    raw0 = torch.cat([warm_start_raw.reshape(1, -1).to(raw0), raw0], 0)

  def loss_fn(raw: torch.Tensor) -> torch.Tensor:
    return negative_log_marginal_likelihood(raw, x, y, linear_coef)

  best_raw, best_f = lbfgs.minimize_batched(loss_fn, raw0,
                                            max_iters=max_iters,
                                            check_every=5)
  idx = int(torch.argmin(best_f))
  raw = best_raw[idx]

  # fp64 cache (same conditioning strategy as gp_model.train_gp).
  x64 = x.double()
  p64 = LinearMaternParams.from_raw(raw.double())
  K = _combined_gram(p64, linear_coef, x64, None)
  noise_eff = torch.maximum(p64.noise, 1e-3 * p64.amplitude ** 2)
  K = K + noise_eff * torch.eye(n, dtype=x64.dtype, device=x.device)
  L64 = cholesky_with_jitter(K, p64.amplitude ** 2)
  resid = (y.double() - p64.mean).unsqueeze(-1)
  z = torch.linalg.solve_triangular(L64, resid, upper=False)
  alpha = torch.linalg.solve_triangular(L64.mT, z, upper=True).squeeze(-1)
  params = LinearMaternParams.from_raw(raw)
  return LinearMaternPosterior(
      x=x, params=params, linear_coef=linear_coef, L=L64.to(x.dtype),
      alpha=alpha.to(x.dtype), nll=float(best_f[idx]),
      raw=raw.detach(),
      aux=torch.stack([params.slope.reshape(()),
                       params.shift.reshape(())]))
