"""Gaussian-process training and posterior prediction (PyTorch + HIP).

Numeric spec matches the reference's VizierGaussianProcess
(vizier/_src/jax/models/tuned_gp_models.py:78): Matern-5/2 ARD kernel
with per-dimension lengthscales, log-uniform initialization inside
SoftClip-style bounds (amplitude [1e-3,10], noise [1e-10,1],
lengthscales [1e-2,1e2]), constant mean, marginal-likelihood fit with
multi-restart L-BFGS (vizier/_src/jax/optimizers/jaxopt_wrappers.py),
retrying-Cholesky jitter escalation (tuned_gp_models.py:92), and a
posterior predictive cache (stochastic_process_model.py:968-997).

MI355X design notes: all restarts are batched into one tensor program;
the posterior additionally precomputes K^-1 so the acquisition sweep's
variance is a GEMM quadform (k @ K^-1 vs per-candidate triangular
solves, which are latency-bound on 64-wide wavefronts).
"""

from __future__ import annotations

import dataclasses
import math
import os
from typing import Optional, Tuple

import torch

from vizier_amd._src.gp import lbfgs
from vizier_amd._src.gp.matern import gram_matern52
from vizier_amd._src.ops import dispatch as ops

# ROCm 7.2 / gfx950: MAGMA's batched Cholesky (the default backend for
# batched inputs) hits hipErrorLaunchFailure for batch > 1 when
# 257 <= N <= ~320 (mapped in profiles/cholprobe2.log; 256 and 384+ are
# fine). Pinning hipSOLVER instead costs 4x on every warm ARD fit
# (profiles/fitprobe.log: 784 vs 201 ms), so we keep the fast default
# backend and identity-pad batched factorizations through the crash
# window — block-diagonal padding is exact: chol([[K,0],[0,I]]) =
# [[chol(K),0],[0,I]], logdet unchanged, and we slice the result back.
_MAGMA_BAD_LO, _MAGMA_BAD_HI = 257, 511  # generous upper margin
_NO_GRAD_FIT_N = 8000  # above this, analytic-gradient fit (see train_gp)
_MAGMA_PAD_N = 512


def safe_cholesky_ex(K: torch.Tensor):
  """torch.linalg.cholesky_ex that avoids MAGMA's batched crash window."""
  n = K.shape[-1]
  if (K.is_cuda and K.dim() > 2 and K.shape[0] > 1 and
      _MAGMA_BAD_LO <= n <= _MAGMA_BAD_HI):
    shape = K.shape[:-2] + (_MAGMA_PAD_N, _MAGMA_PAD_N)
    Kp = torch.zeros(shape, dtype=K.dtype, device=K.device)
    Kp[..., :n, :n] = K
    idx = torch.arange(n, _MAGMA_PAD_N, device=K.device)
    Kp[..., idx, idx] = 1.0
    L, info = torch.linalg.cholesky_ex(Kp)
    return L[..., :n, :n].contiguous(), info
  return torch.linalg.cholesky_ex(K)

# SoftClip-style bounds (log-space), mirroring tuned_gp_models.py:147-199.
_LOG_AMP_BOUNDS = (math.log(1e-3), math.log(10.0))
_LOG_NOISE_BOUNDS = (math.log(1e-10), math.log(1.0))
_LOG_LS_BOUNDS = (math.log(1e-2), math.log(1e2))
_MEAN_BOUNDS = (-3.0, 3.0)


def _to_bounded(raw: torch.Tensor, lo: float, hi: float) -> torch.Tensor:
  return lo + (hi - lo) * torch.sigmoid(raw)


def _from_bounded(v: float, lo: float, hi: float) -> float:
  u = (v - lo) / (hi - lo)
  u = min(max(u, 1e-6), 1 - 1e-6)
  return math.log(u / (1 - u))


@dataclasses.dataclass
class GPParams:
  """Constrained GP hyperparameters."""

  amplitude: torch.Tensor      # (...,)
  noise: torch.Tensor          # (...,) observation noise variance
  lengthscales: torch.Tensor   # (..., D)
  mean: torch.Tensor           # (...,)

  @classmethod
  def from_raw(cls, raw: torch.Tensor) -> 'GPParams':
    """raw: (..., D+3) unconstrained -> bounded params."""
    log_amp = _to_bounded(raw[..., 0], *_LOG_AMP_BOUNDS)
    log_noise = _to_bounded(raw[..., 1], *_LOG_NOISE_BOUNDS)
    mean = _to_bounded(raw[..., 2], *_MEAN_BOUNDS)
    log_ls = _to_bounded(raw[..., 3:], *_LOG_LS_BOUNDS)
    return cls(amplitude=log_amp.exp(), noise=log_noise.exp(),
               lengthscales=log_ls.exp(), mean=mean)


def _init_raw(num_restarts: int, dim: int, generator: torch.Generator,
              device, dtype) -> torch.Tensor:
  """Log-uniform init inside the bounds == uniform in sigmoid space."""
  u = torch.rand(num_restarts, dim + 3, generator=generator,
                 device=device, dtype=dtype) * 0.9 + 0.05
  raw = torch.log(u / (1 - u))
  # First restart: a sane default (amp 1, noise 1e-4, ls 1, mean 0).
  raw[0, 0] = _from_bounded(math.log(1.0), *_LOG_AMP_BOUNDS)
  raw[0, 1] = _from_bounded(math.log(1e-4), *_LOG_NOISE_BOUNDS)
  raw[0, 2] = _from_bounded(0.0, *_MEAN_BOUNDS)
  raw[0, 3:] = _from_bounded(math.log(0.5), *_LOG_LS_BOUNDS)
  return raw


def _blocked_solve_lower(L: torch.Tensor, B: torch.Tensor,
                         nb: int = 2048) -> torch.Tensor:
  """Solves L Z = B (L lower-triangular) by blocked forward
  substitution: (nb, nb) diagonal trsm solves + GEMM updates. hipBLAS's
  trsm fails with ALLOC_FAILED whenever lda is ~10000 on ROCm 7.2
  (any rhs width > 1), so huge solves must never hand it the full L."""
  n = L.shape[-1]
  ncols = B.shape[-1]
  Z = torch.empty_like(B)
  # Panel BOTH dimensions: hipBLAS trsm ALLOC_FAILs whenever either
  # extent is ~10^4, so every call here is at most (nb, nb).
  for c0 in range(0, ncols, nb):
    c1 = min(c0 + nb, ncols)
    for i0 in range(0, n, nb):
      i1 = min(i0 + nb, n)
      rhs = B[i0:i1, c0:c1].clone()
      if i0 > 0:
        rhs -= L[i0:i1, :i0] @ Z[:i0, c0:c1]
      Z[i0:i1, c0:c1] = torch.linalg.solve_triangular(
          L[i0:i1, i0:i1].contiguous(), rhs, upper=False)
  return Z


def _chol_solve(L: torch.Tensor, b: torch.Tensor) -> torch.Tensor:
  """(L L^T)^-1 b via two triangular solves (batched)."""
  z = torch.linalg.solve_triangular(L, b, upper=False)
  return torch.linalg.solve_triangular(L.mT, z, upper=True)


def cholesky_with_jitter(K: torch.Tensor, amplitude2: torch.Tensor,
                         max_tries: int = 6) -> torch.Tensor:
  """Batched Cholesky with escalating jitter (tuned_gp_models.py:92)."""
  jitter = 1e-6
  eye = torch.eye(K.shape[-1], dtype=K.dtype, device=K.device)
  for _ in range(max_tries):
    L, info = safe_cholesky_ex(K)
    if bool((info == 0).all()):
      return L
    scale = amplitude2.reshape(amplitude2.shape + (1, 1)) \
        if amplitude2.dim() > 0 else amplitude2
    bad = (info != 0).reshape(info.shape + (1, 1)).to(K.dtype)
    K = K + bad * jitter * scale * eye
    jitter *= 10.0
  # Last resort: return whatever factorization we can get.
  return safe_cholesky_ex(K)[0]


def _use_custom_chol(K: torch.Tensor) -> bool:
  """Opt-in custom batched potrf/trsv on the no-grad GPU path.

  DEFAULT ('both'): the panel-swept v2 potrf (batched_chol.hip —
  per-matrix panel factor + chip-filling (R x column-tiles) trailing
  kernels, 2 launches per 32-panel) plus the batched trsv replace
  MAGMA's ~5000 spotf2 panel launches and the serial rocblas trsv
  dispatch on the no-grad line-search path. A/B at the headline shape:
  fit 153.6 -> 150.0 ms, steady suggest 350 -> 340 ms. (The v1
  one-workgroup-per-matrix factorization measured SLOWER — 16 CUs
  busy, latency-chained panels — and was replaced by v2.) Opt out with
  VIZIER_AMD_CUSTOM_CHOL=0 / trsv-only with =trsv.
  """
  mode = os.environ.get('VIZIER_AMD_CUSTOM_CHOL', 'both')
  if mode not in ('1', 'both'):
    return False
  # N <= 1024 only: at N=2000 the 63 launch-serial panel rounds lose
  # to MAGMA (measured 11 ms/call vs ~4, profiles/q3_kernels_r2.txt);
  # at N=1000 v2 wins (fit 153.6 -> 150 ms).
  return (K.is_cuda and K.dtype == torch.float32 and
          not K.requires_grad and K.shape[0] > 1 and
          K.shape[-1] <= 1024 and ops.extension_available())


def _use_custom_trsv(K: torch.Tensor) -> bool:
  """MAGMA factorization + the batched wave-synchronous solve: applies
  whenever the custom solve is enabled (any non-'0' mode) but the full
  custom factorization is not. N <= 1200: the one-workgroup-per-matrix
  solve wins at N=1000 (0.48 vs 0.71 ms rocblas) but LOSES at N=2000
  (3.1 vs 1.5 ms — 63 serial panel rounds on one CU;
  tools_chol_backends.py probe)."""
  if os.environ.get('VIZIER_AMD_CUSTOM_CHOL', 'both') not in (
      'trsv', 'both', '1'):
    return False
  return (K.is_cuda and K.dtype == torch.float32 and
          not K.requires_grad and K.shape[0] > 1 and
          K.shape[-1] <= 1200 and ops.extension_available())


def nll_values_with_chol(raw: torch.Tensor, x: torch.Tensor,
                         y: torch.Tensor):
  """Batched NLL that also returns its Cholesky factors.

  Returns (nll (R,), L (R, N, N), info (R,) int). The factors let the
  L-BFGS line search hand the ACCEPTED candidate's factorization to
  the gradient evaluation (lbfgs.py `ladder_fn`), skipping one full
  batched potrf per iteration — x_new is bit-identical to the selected
  ladder row, so L here factors the same K up to the last-ulp
  difference between this K builder (gram_matern52) and the gradient
  eval's explicit-distance build."""
  params = GPParams.from_raw(raw)
  n = x.shape[0]
  if (x.is_cuda and x.dtype == torch.float32 and not raw.requires_grad
      and ops.extension_available()):
    # One fused kernel builds K for every restart with noise*I folded
    # in, replacing the 7-pass torch chain (scale, GEMM d2, clamp,
    # sqrt, exp, mul, add-eye) over (R, N, N).
    K = ops.require_ext().gram_matern52_batched(
        x.contiguous(), params.lengthscales.contiguous(),
        params.amplitude.contiguous(), params.noise.contiguous(),
        False)[0]
  else:
    K = gram_matern52(x.unsqueeze(0), None, params.lengthscales,
                      params.amplitude)
    noise = params.noise.reshape(-1, 1, 1)
    K = K + noise * torch.eye(n, dtype=x.dtype, device=x.device)
  resid = (y.unsqueeze(0) - params.mean.unsqueeze(-1)).unsqueeze(-1)
  if _use_custom_chol(K):
    ext = ops.require_ext()
    L, info = ext.batched_potrf(K.contiguous())
    z = ext.batched_trsv_lower(
        L, resid.squeeze(-1).expand(K.shape[0], n).contiguous())
    quad = (z * z).sum(dim=-1)
  else:
    L, info = safe_cholesky_ex(K)
    if _use_custom_trsv(K):
      ext = ops.require_ext()
      z = ext.batched_trsv_lower(
          L.contiguous(),
          resid.squeeze(-1).expand(K.shape[0], n).contiguous())
      quad = (z * z).sum(dim=-1)
    else:
      # quad = r^T K^-1 r = ||L^-1 r||^2: ONE triangular solve instead
      # of a _chol_solve pair — the batched vector-RHS solves dispatch
      # as serial rocBLAS trsv calls, so halving them matters.
      # (torch.cholesky_solve itself hipErrorLaunchFailures here.)
      z = torch.linalg.solve_triangular(L, resid, upper=False)
      quad = (z * z).sum(dim=(-1, -2))
  logdet = 2.0 * torch.log(torch.diagonal(L, dim1=-2, dim2=-1)).sum(-1)
  nll = 0.5 * (quad + logdet + n * math.log(2 * math.pi))
  # Mild pull toward the raw-space origin (the reference regularizes via
  # its parameter priors); also keeps sigmoid saturation in check.
  nll = nll + 0.01 * (raw * raw).sum(-1)
  # Failed factorizations get +inf so the optimizer backs off.
  nll = torch.where(info == 0, nll, torch.full_like(nll, float('inf')))
  return nll, L, info


def negative_log_marginal_likelihood(raw: torch.Tensor, x: torch.Tensor,
                                     y: torch.Tensor) -> torch.Tensor:
  """Batched NLL over restarts. raw (R, D+3); x (N, D); y (N,)."""
  nll, _, _ = nll_values_with_chol(raw, x, y)
  return nll


def nll_value_and_grad(raw: torch.Tensor, x: torch.Tensor,
                       y: torch.Tensor, chol_hint=None,
                       ) -> Tuple[torch.Tensor, torch.Tensor]:
  """Batched NLL and its ANALYTIC gradient (no autograd).

  Purpose: torch's Cholesky/solve backward needs hipBLAS trsm calls
  that ALLOC_FAIL at N~10^4 on ROCm 7.2 (profiles/config4*.log), so
  huge studies cannot autograd the NLL. This path uses only forward
  factorizations + the classic GP gradient identities:

    dNLL/dtheta = 1/2 tr((K^-1 - alpha alpha^T) dK/dtheta)
                  (+ mean-parameter term  -sum(alpha)),

  with the per-lengthscale traces reduced to GEMV forms
  (sum_ij A_ij (z_id - z_jd)^2 = 2 z_d^2 . rowsum(A) - 2 z_d^T A z_d
  for symmetric A). Verified against autograd in tests/test_gp_core.py.

  raw (R, D+3) -> (nll (R,), grad (R, D+3)).
  """
  r_batch, p = raw.shape
  n, d = x.shape
  params = GPParams.from_raw(raw)
  amp2 = (params.amplitude ** 2).reshape(-1, 1, 1)          # (R,1,1)
  ls = params.lengthscales                                   # (R, D)
  z = x.unsqueeze(0) / ls.unsqueeze(1)                       # (R, N, D)
  fused_gram = (x.is_cuda and x.dtype == torch.float32 and
                ops.extension_available())
  if fused_gram:
    # One kernel emits K (+noise*I) AND the gradient factor
    # G = amp^2 (5/3)(1+sr)e from the same distance pass.
    K, G = ops.require_ext().gram_matern52_batched(
        x.contiguous(), ls.contiguous(),
        params.amplitude.contiguous(), params.noise.contiguous(), True)
    k0 = None
  else:
    d2 = ((z * z).sum(-1, keepdim=True) +
          (z * z).sum(-1).unsqueeze(-2) -
          2.0 * z @ z.transpose(-1, -2)).clamp_min(1e-18)
    r = d2.sqrt()
    sr = math.sqrt(5.0) * r
    e = torch.exp(-sr)
    k0 = (1.0 + sr + sr * sr / 3.0) * e                      # unit-amp
    G = amp2 * (5.0 / 3.0) * (1.0 + sr) * e
    K = amp2 * k0 + params.noise.reshape(-1, 1, 1) * torch.eye(
        n, dtype=x.dtype, device=x.device)
  resid = (y.unsqueeze(0) - params.mean.unsqueeze(-1)).unsqueeze(-1)
  if chol_hint is not None:
    # The line-search ladder already factored K at these exact raw
    # values (bit-identical x_new); reuse its (L, info) and skip the
    # batched potrf entirely.
    L, info = chol_hint
    L = L.contiguous()
    if _use_custom_chol(K) or _use_custom_trsv(K):
      ext = ops.require_ext()
      zs = ext.batched_trsv_lower(
          L, resid.squeeze(-1).expand(K.shape[0], n).contiguous())
      quad = (zs * zs).sum(dim=-1)
    else:
      zsol = torch.linalg.solve_triangular(L, resid, upper=False)
      quad = (zsol * zsol).sum(dim=(-1, -2))
  elif _use_custom_chol(K):
    # Headline-shape fast path: custom panel-swept potrf + the batched
    # wave-synchronous solve instead of MAGMA's spotf2 launch storm
    # (profiles/fit_kernels_headline.txt: the MAGMA+trtri+serial-trsv
    # autograd machinery was ~45 ms of the 157 ms warm refit).
    ext = ops.require_ext()
    L, info = ext.batched_potrf(K.contiguous())
    zs = ext.batched_trsv_lower(
        L, resid.squeeze(-1).expand(K.shape[0], n).contiguous())
    quad = (zs * zs).sum(dim=-1)
  else:
    L, info = safe_cholesky_ex(K)
    zsol = torch.linalg.solve_triangular(L, resid, upper=False)
    quad = (zsol * zsol).sum(dim=(-1, -2))
  logdet = 2.0 * torch.log(torch.diagonal(L, dim1=-2, dim2=-1)).sum(-1)
  nll = 0.5 * (quad + logdet + n * math.log(2 * math.pi))
  nll = nll + 0.01 * (raw * raw).sum(-1)

  # K^-1 via Z = L^-1 (trsm) then W = Z^T Z; alpha = K^-1 r = W r, so
  # no serial upper trsv is needed. Huge N keeps the blocked per-restart
  # solves (hipBLAS trsm ALLOC_FAILs at lda ~10^4); moderate N uses ONE
  # batched trsm + batched GEMM.
  eye = torch.eye(n, dtype=x.dtype, device=x.device)
  if n >= _NO_GRAD_FIT_N:
    W = torch.empty_like(K)
    for i in range(r_batch):
      zi = _blocked_solve_lower(L[i], eye)
      W[i] = zi.T @ zi
  else:
    Z = torch.linalg.solve_triangular(
        L, eye.expand(r_batch, n, n), upper=False)
    W = Z.mT @ Z
  alpha = W @ resid                                          # (R, N, 1)
  M = W - alpha @ alpha.mT                                   # (R, N, N)

  grad = torch.zeros_like(raw)
  # d(bounded log-param)/d(raw) for the sigmoid reparameterization.
  sig = torch.sigmoid(raw)
  dbound = sig * (1 - sig)

  # amplitude (raw col 0): dK/dv = 2 * amp^2 * k0, v = log amp.
  tr_m = torch.diagonal(M, dim1=-2, dim2=-1).sum(-1)         # (R,)
  if k0 is not None:
    g_amp = 0.5 * (M * (2.0 * amp2 * k0)).sum(dim=(-1, -2))
  else:
    # amp^2 k0 == K - noise*I, so the trace rewrites without k0:
    # 0.5 tr(M * 2(K - noise I)) = sum(M*K) - noise * tr(M).
    g_amp = (M * K).sum(dim=(-1, -2)) - params.noise * tr_m
  grad[:, 0] = g_amp * (_LOG_AMP_BOUNDS[1] - _LOG_AMP_BOUNDS[0]) *       dbound[:, 0]
  # noise (col 1): dK/dv = noise * I.
  g_noise = 0.5 * params.noise * tr_m
  grad[:, 1] = g_noise * (_LOG_NOISE_BOUNDS[1] -
                          _LOG_NOISE_BOUNDS[0]) * dbound[:, 1]
  # mean (col 2): dNLL/dm = -sum(alpha).
  g_mean = -alpha.sum(dim=(-1, -2))
  grad[:, 2] = g_mean * (_MEAN_BOUNDS[1] - _MEAN_BOUNDS[0]) *       dbound[:, 2]
  # lengthscales (cols 3:): A = 1/2 M * G with
  # G = amp^2 * (5/3)(1 + sqrt5 r) e^{-sqrt5 r}; symmetric. All D
  # traces in ONE batched GEMM: z_d^T A z_d = sum_i z_id (A z)_id and
  # sum_ij A_ij z_id^2 = sum_i z_id^2 rowsum(A)_i (a per-d einsum loop
  # re-read the (R, N, N) A tensor D times).
  A = 0.5 * M * G
  arow = A.sum(-1)                                           # (R, N)
  az = torch.bmm(A, z)                                       # (R, N, D)
  t_all = 2.0 * ((z * z) * arow.unsqueeze(-1)).sum(dim=1) - \
      2.0 * (z * az).sum(dim=1)                              # (R, D)
  grad[:, 3:] = t_all * (_LOG_LS_BOUNDS[1] - _LOG_LS_BOUNDS[0]) * \
      dbound[:, 3:]

  grad = grad + 0.02 * raw
  bad = info != 0
  nll = torch.where(bad, torch.full_like(nll, float('inf')), nll)
  grad = torch.where(bad.unsqueeze(-1), torch.zeros_like(grad), grad)
  return nll, grad


@dataclasses.dataclass
class GPPosterior:
  """Cached posterior state for fast repeated prediction."""

  x: torch.Tensor            # (N, D) training features
  params: GPParams           # scalar-valued (best restart)
  L: torch.Tensor            # (N, N) Cholesky of K + noise I
  alpha: torch.Tensor        # (N,) (K + noise I)^-1 (y - mean)
  K_inv: Optional[torch.Tensor]  # (N, N), for the GEMM variance path
  nll: float                 # training loss of the selected restart
  raw: Optional[torch.Tensor] = None  # unconstrained params (warm starts)
  noise_eff: float = 0.0     # cache noise floor actually used (see train_gp)

  def predict(self, xq: torch.Tensor) -> Tuple[torch.Tensor, torch.Tensor]:
    """Returns (mean, stddev) of the latent f at query points (Q, D)."""
    k = ops.gram_matern52(xq, self.x, self.params.lengthscales,
                          self.params.amplitude)           # (Q, N)
    mean = self.params.mean + k @ self.alpha
    amp2 = self.params.amplitude ** 2
    if self.K_inv is not None:
      var = amp2 - (k * (k @ self.K_inv)).sum(-1)
    else:
      v = torch.linalg.solve_triangular(self.L, k.T, upper=False)
      var = amp2 - (v * v).sum(0)
    return mean, var.clamp_min(1e-12).sqrt()

  def predict_cov(self, xq: torch.Tensor
                  ) -> Tuple[torch.Tensor, torch.Tensor]:
    """Returns (mean, full covariance) for a small query batch (q-EI)."""
    k = ops.gram_matern52(xq, self.x, self.params.lengthscales,
                          self.params.amplitude)           # (Q, N)
    mean = self.params.mean + k @ self.alpha
    k_qq = ops.gram_matern52(xq, xq, self.params.lengthscales,
                             self.params.amplitude)
    if self.K_inv is not None:
      cov = k_qq - k @ self.K_inv @ k.T
    else:
      v = torch.linalg.solve_triangular(self.L, k.T, upper=False)
      cov = k_qq - v.T @ v
    return mean, cov


@dataclasses.dataclass
class EnsembleGPPosterior:
  """Equal-weight mixture of the best-N ARD restarts' posteriors.

  Parity with UniformEnsemblePredictive
  (vizier/_src/jax/stochastic_process_model.py:835) selected by
  ensemble_size (gp_models.py:201): predictions are a uniform Gaussian
  mixture over members — mean is the member-mean average, variance is
  E[m^2 + s^2] - E[m]^2. Members are sorted by NLL, so .params/.raw/
  .nll expose the best member (warm starts keep working).
  """

  members: list

  @property
  def x(self) -> torch.Tensor:
    return self.members[0].x

  @property
  def params(self) -> GPParams:
    return self.members[0].params

  @property
  def nll(self) -> float:
    return self.members[0].nll

  @property
  def raw(self) -> Optional[torch.Tensor]:
    return self.members[0].raw

  @property
  def noise_eff(self) -> float:
    return self.members[0].noise_eff

  @property
  def K_inv(self):
    # None: the fused K^-1 quadform scorers apply to a single GP;
    # ScoringFunction then falls back to predict() (the mixture).
    return None

  def predict(self, xq: torch.Tensor) -> Tuple[torch.Tensor, torch.Tensor]:
    means, stds = zip(*(m.predict(xq) for m in self.members))
    means = torch.stack(means)
    stds = torch.stack(stds)
    mix_mean = means.mean(0)
    mix_var = (stds.square() + means.square()).mean(0) - mix_mean.square()
    return mix_mean, mix_var.clamp_min(1e-12).sqrt()


def train_gp(x: torch.Tensor, y: torch.Tensor, *,
             num_restarts: int = 4, max_iters: int = 50,
             seed: int = 0, precompute_inverse: bool = True,
             warm_start_raw: Optional[torch.Tensor] = None,
             ensemble_size: int = 1):
  """Fits GP hyperparameters by restarting batched L-BFGS on the NLL.

  Mirrors gp_models.train_gp (vizier/_src/algorithms/designers/gp/
  gp_models.py:169-223): restart init -> ARD optimize -> best restart ->
  posterior precompute. With ensemble_size > 1 (gp_models.py:201) the
  best N finite restarts are kept and returned as an
  EnsembleGPPosterior (uniform mixture), else a single GPPosterior.
  """
  x = x.detach()
  y = y.detach().reshape(-1)
  n, d = x.shape
  generator = torch.Generator(device='cpu').manual_seed(seed)
  raw0 = _init_raw(num_restarts + 1, d, generator, 'cpu',
                   torch.float32).to(device=x.device, dtype=x.dtype)
  if warm_start_raw is not None and warm_start_raw.numel() == d + 3:
    # Warm start from the previous fit's optimum (incremental refits).
    # NOTE r2: tried shrinking the warm restart batch / line-search
    # ladder for the 166 ms refit — small-N convergence regressed
    # (tests/test_gp_bandit convergence gates), so the regret-validated
    # r1 schedule stands.
    raw0 = torch.cat([warm_start_raw.reshape(1, -1).to(raw0), raw0], 0)

  def loss_fn(raw: torch.Tensor) -> torch.Tensor:
    return negative_log_marginal_likelihood(raw, x, y)

  # VIZIER_AMD_ANALYTIC_NLL: 'auto' (default), 'always', or 'never'.
  # 'auto' uses analytic gradients (a) at N >= _NO_GRAD_FIT_N where
  # torch's trsm-backward is broken, and (b) on the GPU fp32 path where
  # the custom potrf/trsv kernels apply — there the analytic grad does
  # ONE batched trsm + GEMM instead of autograd's MAGMA spotf2 +
  # trtri/trsm backward + serial rocblas trsv chain (warm refit
  # 146.8 -> 132.9 ms, headline bench 343 -> 324.7 ms; profiles/
  # fit_kernels_headline.txt). Gradients are the exact trace-identity
  # values, verified against autograd in tests/test_gp_core.py.
  analytic_mode = os.environ.get('VIZIER_AMD_ANALYTIC_NLL', 'auto')
  use_analytic = (analytic_mode == 'always' or
                  (analytic_mode != 'never' and
                   (n >= _NO_GRAD_FIT_N or
                    (x.is_cuda and x.dtype == torch.float32 and
                     ops.extension_available()))))
  if max_iters <= 0:
    with torch.no_grad():
      f0 = loss_fn(raw0)
    best_raw, best_f = raw0, f0
  elif use_analytic:
    # hipBLAS's trsm BACKWARD fails with ALLOC_FAILED at N~10^4 on
    # ROCm 7.2 (even unbatched), so huge studies cannot autograd the
    # NLL. Fit with ANALYTIC gradients instead (forward-only solves,
    # verified against autograd to 1e-9 in tests/test_gp_core.py).
    # Ladder-cache: the line search returns its (L, info) per trial so
    # the gradient eval at the accepted point (bit-identical raw)
    # skips one full batched potrf per iteration. Gated on cache size
    # (the (S*R, N, N) factors are ~6 GB at config-4 scale).
    r_ladder = raw0.shape[0] * 4  # ls_steps ladder width
    ladder = None
    if r_ladder * n * n * 4 <= (1 << 30):
      def ladder(raw):
        nll_v, L, information = nll_values_with_chol(raw, x, y)
        return nll_v, (L, information)
    best_raw, best_f = lbfgs.minimize_batched(
        loss_fn, raw0, max_iters=max_iters, check_every=5,
        value_and_grad_fn=lambda raw, hint=None: nll_value_and_grad(
            raw, x, y, chol_hint=hint),
        ladder_fn=ladder)
  else:
    best_raw, best_f = lbfgs.minimize_batched(loss_fn, raw0,
                                              max_iters=max_iters,
                                              check_every=5)
  def _build_posterior(raw: torch.Tensor, nll_one: float) -> GPPosterior:
    return _build_posterior_cache(
        x, y, raw, nll_one, precompute_inverse=precompute_inverse)

  order = torch.argsort(best_f)
  if ensemble_size <= 1:
    i = int(order[0])
    return _build_posterior(best_raw[i], float(best_f[i]))
  members = []
  for i in order[:ensemble_size].tolist():
    if math.isfinite(float(best_f[i])):
      members.append(_build_posterior(best_raw[i], float(best_f[i])))
  if len(members) <= 1:
    i = int(order[0])
    return members[0] if members else _build_posterior(
        best_raw[i], float(best_f[i]))
  return EnsembleGPPosterior(members=members)


def _build_posterior_cache(x: torch.Tensor, y: torch.Tensor,
                           raw: torch.Tensor, nll_one: float, *,
                           precompute_inverse: bool) -> GPPosterior:
  """Builds the fp64 posterior cache for one raw-parameter vector."""
  n = x.shape[0]
  params = GPParams.from_raw(raw)

  # Posterior caches in fp64, stored fp32: once the fit drives noise
  # toward its lower bound, cond(K) ~ amp^2/noise exceeds fp32's
  # 1/eps and the explicit-inverse quadform (the HIP sweep's variance
  # path) loses all accuracy. One fp64 factorization per fit is
  # microseconds-scale on MI355X (CDNA4 FP64 matrix cores) and makes
  # the cached alpha / K_inv correct to fp32 resolution.
  x64 = x.double()
  p64 = GPParams.from_raw(raw.double())
  K = gram_matern52(x64, None, p64.lengthscales, p64.amplitude)
  # fp32-consumable conditioning floor: the NLL fit may drive noise to
  # ~1e-10 (the reference runs in float64 — jax x64), but the cached
  # K_inv feeds fp32 GEMM quadforms whose cancellation error scales as
  # eps * amp^4 / noise. Flooring the CACHE's noise at 1e-3*amp^2
  # bounds that error to ~6e-5*amp^2 (a ~3% stddev error at the floor)
  # at the cost of a ~0.03*amp stddev floor near training points —
  # negligible against UCB-scale acquisition scores.
  # The floor only matters when the cache feeds the fp32 explicit-inverse
  # quadform; the triangular-solve (composed/predict) path is fp64-backed
  # and matches reference variances better with the true fitted noise
  # (reference lower bound is 1e-10 in float64).
  if precompute_inverse:
    noise_eff = torch.maximum(p64.noise, 1e-3 * p64.amplitude ** 2)
  else:
    noise_eff = p64.noise
  K = K + noise_eff * torch.eye(n, dtype=x64.dtype, device=x.device)
  L64 = cholesky_with_jitter(K, p64.amplitude ** 2)
  resid = (y.double() - p64.mean).unsqueeze(-1)
  alpha = _chol_solve(L64, resid).squeeze(-1)
  K_inv = None
  if precompute_inverse:
    eye = torch.eye(n, dtype=x64.dtype, device=x.device)
    if n >= _NO_GRAD_FIT_N:
      z = _blocked_solve_lower(L64, eye)
    else:
      z = torch.linalg.solve_triangular(L64, eye, upper=False)
    K_inv = (z.T @ z).to(x.dtype)
  return GPPosterior(x=x, params=params, L=L64.to(x.dtype),
                     alpha=alpha.to(x.dtype), K_inv=K_inv,
                     nll=nll_one, raw=raw.detach(),
                     noise_eff=float(noise_eff))
