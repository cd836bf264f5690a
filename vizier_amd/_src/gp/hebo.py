"""HEBO-style alternative GP surrogate.

Capability parity with vizier/_src/jax/models/hebo_gp_model.py
(VizierHeboGaussianProcess :41): zero-mean GP with

  k(x, x') = signal_variance * Matern-3/2(||(w(x) - w(x')) / ls||)
             + <w(x), w(x')>                      (Linear kernel)

where w is the per-dimension Kumaraswamy CDF input warp
w(x) = 1 - (1 - x^c1)^c0 with concentrations in (0, 10), and the
hyperparameters carry the reference's priors as MAP regularizers:
signal_variance ~ Gamma(0.5, 1), noise ~ LogNormal(-4.63, 0.5),
lengthscales ~ LogNormal(0, 1), c0/c1 ~ LogNormal(0, 0.75).

Training reuses the batched sync-free L-BFGS machinery (gp/lbfgs.py);
the posterior is Cholesky-backed (this model is an alternative
surrogate, not the HIP sweep hot path).
"""

from __future__ import annotations

import dataclasses
import math
from typing import Optional, Tuple

import torch

from vizier_amd._src.gp import gp_model, lbfgs

_SQRT3 = math.sqrt(3.0)
_EPS = 1e-6


def _softplus(x: torch.Tensor) -> torch.Tensor:
  return torch.nn.functional.softplus(x) + _EPS


def matern32(r: torch.Tensor) -> torch.Tensor:
  sr = _SQRT3 * r
  return (1.0 + sr) * torch.exp(-sr)


def kumaraswamy_warp(x: torch.Tensor, c1: torch.Tensor,
                     c0: torch.Tensor) -> torch.Tensor:
  """CDF warp on [0,1]^D; c1/c0 broadcast over the feature axis."""
  x = x.clamp(1e-6, 1 - 1e-6)
  return 1.0 - (1.0 - x ** c1) ** c0


@dataclasses.dataclass
class HeboParams:
  signal_variance: torch.Tensor   # (...,)
  noise: torch.Tensor             # (...,)
  lengthscales: torch.Tensor      # (..., D)
  c0: torch.Tensor                # (...,) Kumaraswamy concentration0
  c1: torch.Tensor                # (...,) concentration1

  @classmethod
  def from_raw(cls, raw: torch.Tensor, d: int) -> 'HeboParams':
    sv = _softplus(raw[..., 0])
    noise = _softplus(raw[..., 1])
    ls = raw[..., 2:2 + d].exp()
    sigmoid10 = lambda t: 10.0 * torch.sigmoid(t)
    return cls(signal_variance=sv, noise=noise, lengthscales=ls,
               c0=sigmoid10(raw[..., 2 + d]).clamp_min(1e-3),
               c1=sigmoid10(raw[..., 3 + d]).clamp_min(1e-3))


def _log_normal_logpdf(v: torch.Tensor, loc: float, scale: float
                       ) -> torch.Tensor:
  lv = v.clamp_min(1e-20).log()
  return (-0.5 * ((lv - loc) / scale) ** 2 - lv -
          math.log(scale * math.sqrt(2 * math.pi)))


def _gamma_half_logpdf(v: torch.Tensor) -> torch.Tensor:
  # Gamma(concentration=0.5, rate=1): log p = -0.5*log(v) - v + const.
  return -0.5 * v.clamp_min(1e-20).log() - v


def _hebo_gram(params: HeboParams, x1: torch.Tensor,
               x2: Optional[torch.Tensor]) -> torch.Tensor:
  """Batched Gram: (..., N, M). Inputs (N, D) shared across the batch."""
  c1 = params.c1.reshape(params.c1.shape + (1, 1))
  c0 = params.c0.reshape(params.c0.shape + (1, 1))
  w1 = kumaraswamy_warp(x1, c1, c0)               # (..., N, D)
  w2 = w1 if x2 is None else kumaraswamy_warp(x2, c1, c0)
  ls = params.lengthscales.unsqueeze(-2)
  z1, z2 = w1 / ls, w2 / ls
  d2 = ((z1 * z1).sum(-1, keepdim=True) +
        (z2 * z2).sum(-1).unsqueeze(-2) -
        2.0 * z1 @ z2.transpose(-1, -2)).clamp_min(1e-18)
  sv = params.signal_variance.reshape(params.signal_variance.shape +
                                      (1, 1))
  k = sv * matern32(d2.sqrt())
  k = k + w1 @ w2.transpose(-1, -2)               # Linear kernel term
  return k


def negative_log_posterior(raw: torch.Tensor, x: torch.Tensor,
                           y: torch.Tensor) -> torch.Tensor:
  """Batched NLL + negative log prior (MAP). raw (R, D+4)."""
  n, d = x.shape
  params = HeboParams.from_raw(raw, d)
  K = _hebo_gram(params, x.unsqueeze(0), None)
  K = K + params.noise.reshape(-1, 1, 1) * torch.eye(
      n, dtype=x.dtype, device=x.device)
  L, info = gp_model.safe_cholesky_ex(K)
  resid = y.reshape(1, n, 1).expand(raw.shape[0], n, 1)
  z = torch.linalg.solve_triangular(L, resid, upper=False)
  quad = (z * z).sum(dim=(-1, -2))
  logdet = 2.0 * torch.log(
      torch.diagonal(L, dim1=-2, dim2=-1)).sum(-1)
  nll = 0.5 * (quad + logdet + n * math.log(2 * math.pi))
  log_prior = (_gamma_half_logpdf(params.signal_variance) +
               _log_normal_logpdf(params.noise, -4.63, 0.5) +
               _log_normal_logpdf(params.lengthscales, 0.0, 1.0).sum(-1) +
               _log_normal_logpdf(params.c0, 0.0, 0.75) +
               _log_normal_logpdf(params.c1, 0.0, 0.75))
  nlp = nll - log_prior
  return torch.where(info == 0, nlp, torch.full_like(nlp, float('inf')))


@dataclasses.dataclass
class HeboPosterior:
  x: torch.Tensor
  y_mean: float
  y_std: float
  params: HeboParams
  L: torch.Tensor
  alpha: torch.Tensor
  nll: float
  raw: Optional[torch.Tensor] = None

  def predict(self, xq: torch.Tensor) -> Tuple[torch.Tensor, torch.Tensor]:
    k = _hebo_gram(self.params, xq, self.x)        # (Q, N)
    mean = k @ self.alpha
    k_qq = _hebo_gram(self.params, xq, xq)
    v = torch.linalg.solve_triangular(self.L, k.T, upper=False)
    var = (torch.diagonal(k_qq) - (v * v).sum(0)).clamp_min(1e-12)
    return (mean * self.y_std + self.y_mean,
            var.sqrt() * self.y_std)


def train_hebo_gp(x: torch.Tensor, y: torch.Tensor, *,
                  num_restarts: int = 4, max_iters: int = 50,
                  seed: int = 0,
                  warm_start_raw: Optional[torch.Tensor] = None
                  ) -> HeboPosterior:
  """MAP fit of the HEBO GP by batched restarted L-BFGS.

  Labels are standardized internally (the model is zero-mean)."""
  x = x.detach()
  y = y.detach().reshape(-1)
  y_mean = float(y.mean())
  y_std = float(y.std().clamp_min(1e-8))
  yn = (y - y_mean) / y_std
  n, d = x.shape
  p = d + 4
  g = torch.Generator(device='cpu').manual_seed(seed)
  raw0 = torch.randn(num_restarts + 1, p, generator=g) * 0.7
  raw0[0] = 0.0
  raw0[0, 1] = -4.0   # near the noise prior mode
  raw0 = raw0.to(device=x.device, dtype=x.dtype)
  if warm_start_raw is not None and warm_start_raw.numel() == p:
    raw0 = torch.cat([warm_start_raw.reshape(1, -1).to(raw0), raw0], 0)

  def loss_fn(raw: torch.Tensor) -> torch.Tensor:
    return negative_log_posterior(raw, x, yn)

  best_raw, best_f = lbfgs.minimize_batched(loss_fn, raw0,
                                            max_iters=max_iters,
                                            check_every=5)
  idx = int(torch.argmin(best_f))
  raw = best_raw[idx]
  params = HeboParams.from_raw(raw, d)
  K = _hebo_gram(HeboParams.from_raw(raw.unsqueeze(0), d),
                 x.unsqueeze(0), None)[0]
  noise_eff = torch.maximum(
      params.noise, 1e-4 * (params.signal_variance + 1.0))
  K = K + noise_eff * torch.eye(n, dtype=x.dtype, device=x.device)
  L = gp_model.cholesky_with_jitter(K, params.signal_variance)
  alpha = gp_model._chol_solve(L, yn.unsqueeze(-1)).squeeze(-1)
  return HeboPosterior(x=x, y_mean=y_mean, y_std=y_std, params=params,
                       L=L, alpha=alpha, nll=float(best_f[idx]),
                       raw=raw.detach())
