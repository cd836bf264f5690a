"""Multitask GPs: INDEPENDENT and SEPARABLE task kernels.

Capability parity with vizier/_src/jax/models/multitask_tuned_gp_models
.py (MultiTaskType :41-59, build_task_kernel_scale_linop :62) and
tuned_gp_models.py:287-295: the SEPARABLE model's joint covariance over
(point, task) pairs is the Kronecker product

    Cov[(i, t), (j, s)] = k_x(x_i, x_j) * B[t, s] + noise * 1[(i,t)=(j,s)]

with a shared Matern-5/2 ARD base kernel k_x (unit amplitude — the task
covariance B carries the scale) and a learned task covariance:

- SEPARABLE ('full'): B = (S Lc)(S Lc)^T where Lc is a row-normalized
  unit-diagonal lower-triangular correlation Cholesky (the reference's
  CorrelationCholesky bijector) and S a bounded per-task scale diag.
- SEPARABLE_DIAG ('diag'): B diagonal (no cross-task correlation).
- INDEPENDENT: one single-task GP per column (gp_model.train_gp).

All restarts are batched into one tensor program (MI355X-first, like
gp_model); the joint Cholesky is (N*M, N*M), fine for the moderate-N
multi-metric regime this model serves.
"""

from __future__ import annotations

import dataclasses
import enum
import math
from typing import List, Optional, Tuple

import torch

from vizier_amd._src.gp import gp_model, lbfgs
from vizier_amd._src.gp.matern import gram_matern52


class MultiTaskType(enum.Enum):
  INDEPENDENT = 'independent'
  SEPARABLE = 'separable'          # full task covariance (LKJ-style)
  SEPARABLE_DIAG = 'separable_diag'

_LOG_TASK_SCALE_BOUNDS = (math.log(1e-3), math.log(10.0))


def _n_task_params(m: int, kind: MultiTaskType) -> int:
  if kind == MultiTaskType.SEPARABLE:
    return m + m * (m - 1) // 2
  return m


def _task_chol(raw_task: torch.Tensor, m: int, kind: MultiTaskType
               ) -> torch.Tensor:
  """raw (..., P) -> lower-triangular Cholesky factor of B (..., M, M)."""
  scales = gp_model._to_bounded(
      raw_task[..., :m], *_LOG_TASK_SCALE_BOUNDS).exp()
  if kind == MultiTaskType.SEPARABLE_DIAG:
    return torch.diag_embed(scales)
  # Row-normalized unit-diagonal lower triangle == correlation Cholesky.
  batch = raw_task.shape[:-1]
  lc = torch.zeros(batch + (m, m), dtype=raw_task.dtype,
                   device=raw_task.device)
  eye = torch.eye(m, dtype=raw_task.dtype, device=raw_task.device)
  lc = lc + eye
  rows, cols = torch.tril_indices(m, m, offset=-1)
  if rows.numel():
    lc = lc.clone()
    lc[..., rows, cols] = raw_task[..., m:]
  lc = lc / lc.norm(dim=-1, keepdim=True)
  return scales.unsqueeze(-1) * lc


@dataclasses.dataclass
class MultitaskParams:
  noise: torch.Tensor          # (...,)
  means: torch.Tensor          # (..., M)
  lengthscales: torch.Tensor   # (..., D)
  task_chol: torch.Tensor      # (..., M, M) lower-tri factor of B

  @property
  def task_cov(self) -> torch.Tensor:
    return self.task_chol @ self.task_chol.mT

  @classmethod
  def from_raw(cls, raw: torch.Tensor, d: int, m: int,
               kind: MultiTaskType) -> 'MultitaskParams':
    noise = gp_model._to_bounded(
        raw[..., 0], *gp_model._LOG_NOISE_BOUNDS).exp()
    means = gp_model._to_bounded(
        raw[..., 1:1 + m], *gp_model._MEAN_BOUNDS)
    ls = gp_model._to_bounded(
        raw[..., 1 + m:1 + m + d], *gp_model._LOG_LS_BOUNDS).exp()
    task = _task_chol(raw[..., 1 + m + d:], m, kind)
    return cls(noise=noise, means=means, lengthscales=ls, task_chol=task)


def _joint_cov(params: MultitaskParams, x: torch.Tensor) -> torch.Tensor:
  """(R, N*M, N*M) = kron(Kx, B) + noise I (row-major (i, t) vec)."""
  n = x.shape[0]
  m = params.task_chol.shape[-1]
  kx = gram_matern52(x.unsqueeze(0), None, params.lengthscales,
                     torch.ones_like(params.noise))      # (R, N, N)
  b = params.task_cov                                    # (R, M, M)
  k = torch.einsum('rij,rts->ritjs', kx, b).reshape(-1, n * m, n * m)
  eye = torch.eye(n * m, dtype=x.dtype, device=x.device)
  return k + params.noise.reshape(-1, 1, 1) * eye


def negative_log_marginal_likelihood(raw: torch.Tensor, x: torch.Tensor,
                                     y: torch.Tensor,
                                     kind: MultiTaskType) -> torch.Tensor:
  """Batched joint NLL. raw (R, P); x (N, D); y (N, M)."""
  n, d = x.shape
  m = y.shape[1]
  params = MultitaskParams.from_raw(raw, d, m, kind)
  k = _joint_cov(params, x)
  L, info = gp_model.safe_cholesky_ex(k)
  resid = (y.unsqueeze(0) - params.means.unsqueeze(-2)).reshape(
      raw.shape[0], n * m, 1)
  z = torch.linalg.solve_triangular(L, resid, upper=False)
  quad = (z * z).sum(dim=(-1, -2))
  logdet = 2.0 * torch.log(torch.diagonal(L, dim1=-2, dim2=-1)).sum(-1)
  nll = 0.5 * (quad + logdet + n * m * math.log(2 * math.pi))
  nll = nll + 0.01 * (raw * raw).sum(-1)
  return torch.where(info == 0, nll, torch.full_like(nll, float('inf')))


@dataclasses.dataclass
class MultitaskPosterior:
  """Cached joint posterior; predict() returns per-task marginals."""

  x: torch.Tensor              # (N, D)
  params: MultitaskParams      # best restart (unbatched)
  L: torch.Tensor              # (NM, NM)
  alpha_mat: torch.Tensor      # (N, M) = unvec of K^-1 (y - mean)
  # Optional GEMM-quadform variance cache. OFF by default here: the
  # joint K's conditioning (small noise x near-duplicate rows) makes
  # the explicit-inverse quadform noticeably less accurate in fp32 than
  # the triangular-solve path, and multitask predict is not the sweep
  # hot loop (the single-task HIP scorer is).
  K_inv: Optional[torch.Tensor]
  nll: float
  raw: Optional[torch.Tensor] = None

  def predict(self, xq: torch.Tensor) -> Tuple[torch.Tensor, torch.Tensor]:
    """(Q, D) -> mean (Q, M), stddev (Q, M)."""
    n, m = self.alpha_mat.shape
    b = self.params.task_cov                     # (M, M)
    kx = gram_matern52(xq, self.x, self.params.lengthscales,
                       torch.ones((), dtype=xq.dtype, device=xq.device))
    mean = self.params.means + kx @ self.alpha_mat @ b   # (Q, M)
    if self.K_inv is not None:
      w = self.K_inv.reshape(n, m, n, m)
      t1 = torch.einsum('qj,jskt,qk->qst', kx, w, kx)    # (Q, M, M)
      quad = torch.einsum('ts,qsu,ut->qt', b, t1, b)     # (Q, M)
    else:
      q = xq.shape[0]
      kstar = torch.einsum('qj,ts->qtjs', kx, b).reshape(q * m, n * m)
      v = torch.linalg.solve_triangular(self.L, kstar.T, upper=False)
      quad = (v * v).sum(0).reshape(q, m)
    var = torch.diagonal(b) - quad
    return mean, var.clamp_min(1e-12).sqrt()


def train_multitask_gp(x: torch.Tensor, y: torch.Tensor, *,
                       multitask_type: MultiTaskType =
                       MultiTaskType.SEPARABLE,
                       num_restarts: int = 4, max_iters: int = 50,
                       seed: int = 0, precompute_inverse: bool = False,
                       warm_start_raw: Optional[torch.Tensor] = None
                       ) -> MultitaskPosterior:
  """Fits the separable multitask GP by batched restarted L-BFGS."""
  if multitask_type == MultiTaskType.INDEPENDENT:
    raise ValueError('Use gp_model.train_gp per task for INDEPENDENT.')
  x = x.detach()
  y = y.detach()
  n, d = x.shape
  m = y.shape[1]
  p = 1 + m + d + _n_task_params(m, multitask_type)
  generator = torch.Generator(device='cpu').manual_seed(seed)
  u = torch.rand(num_restarts + 1, p, generator=generator) * 0.9 + 0.05
  raw0 = torch.log(u / (1 - u))
  raw0[:, 1 + m + d + m:] = raw0[:, 1 + m + d + m:] * 0.3  # small corr
  raw0[0, 0] = gp_model._from_bounded(math.log(1e-4),
                                      *gp_model._LOG_NOISE_BOUNDS)
  raw0[0, 1:1 + m] = gp_model._from_bounded(0.0, *gp_model._MEAN_BOUNDS)
  raw0[0, 1 + m:1 + m + d] = gp_model._from_bounded(
      math.log(0.5), *gp_model._LOG_LS_BOUNDS)
  raw0[0, 1 + m + d:1 + m + d + m] = gp_model._from_bounded(
      math.log(1.0), *_LOG_TASK_SCALE_BOUNDS)
  raw0[0, 1 + m + d + m:] = 0.0
  raw0 = raw0.to(device=x.device, dtype=x.dtype)
  if warm_start_raw is not None and warm_start_raw.numel() == p:
    raw0 = torch.cat([warm_start_raw.reshape(1, -1).to(raw0), raw0], 0)

  def loss_fn(raw: torch.Tensor) -> torch.Tensor:
    return negative_log_marginal_likelihood(raw, x, y, multitask_type)

  best_raw, best_f = lbfgs.minimize_batched(loss_fn, raw0,
                                            max_iters=max_iters,
                                            check_every=5)
  idx = int(torch.argmin(best_f))
  raw = best_raw[idx]
  params = MultitaskParams.from_raw(raw, d, m, multitask_type)

  # Same fp32 conditioning floor as gp_model.train_gp's cache build.
  pb = MultitaskParams.from_raw(raw.unsqueeze(0), d, m, multitask_type)
  k = _joint_cov(pb, x)[0]
  floor = (1e-3 * params.task_cov.diagonal().max() - params.noise
           ).clamp_min(0.0)
  k = k + floor * torch.eye(n * m, dtype=x.dtype, device=x.device)
  L = gp_model.cholesky_with_jitter(
      k, params.task_cov.diagonal().max())
  resid = (y - params.means).reshape(n * m, 1)
  alpha = gp_model._chol_solve(L, resid).reshape(n, m)
  k_inv = None
  if precompute_inverse:
    eye = torch.eye(n * m, dtype=x.dtype, device=x.device)
    z = torch.linalg.solve_triangular(L, eye, upper=False)
    k_inv = z.T @ z
  return MultitaskPosterior(x=x, params=params, L=L, alpha_mat=alpha,
                            K_inv=k_inv, nll=float(best_f[idx]),
                            raw=raw.detach())


def train_independent_gps(x: torch.Tensor, y: torch.Tensor, *,
                          num_restarts: int = 4, max_iters: int = 50,
                          seed: int = 0) -> List[gp_model.GPPosterior]:
  """MultiTaskType.INDEPENDENT: one single-task GP per column."""
  return [gp_model.train_gp(x, y[:, t], num_restarts=num_restarts,
                            max_iters=max_iters, seed=seed + t)
          for t in range(y.shape[1])]
