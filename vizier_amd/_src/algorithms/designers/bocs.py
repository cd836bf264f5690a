"""BOCS designer: Bayesian Optimization of Combinatorial Structures.

Capability parity with vizier/_src/algorithms/designers/bocs.py
(_BayesianHorseshoeLinearRegression :38, BOCSDesigner): a sparse
Bayesian linear model over binary quadratic features
(x_i and x_i*x_j) fit with a horseshoe-prior Gibbs sampler, and a
simulated-annealing acquisition over {0,1}^n using a Thompson sample of
the model weights. Boolean search spaces only (arXiv:1806.08838).
"""

from __future__ import annotations

import itertools
from typing import List, Optional, Sequence, Tuple

import numpy as np

from vizier_amd import pyvizier as vz
from vizier_amd._src.algorithms.core.abstractions import (
    ActiveTrials,
    CompletedTrials,
    Designer,
)
from vizier_amd._src.algorithms.designers.harmonica import _bool_configs


class BayesianHorseshoeLinearRegression:
  """Horseshoe-prior BLR via a short Gibbs chain (Makalic & Schmidt '16)."""

  def __init__(self, n_iters: int = 200, seed: Optional[int] = None):
    self.n_iters = n_iters
    self._rng = np.random.default_rng(seed)
    self.beta_samples: Optional[np.ndarray] = None

  def fit(self, X: np.ndarray, y: np.ndarray) -> None:
    n, p = X.shape
    rng = self._rng
    sigma2, tau2 = 1.0, 1.0
    lam2 = np.ones(p)
    nu = np.ones(p)
    xi = 1.0
    XtX = X.T @ X
    Xty = X.T @ y
    betas = []
    beta = np.zeros(p)
    for it in range(self.n_iters):
      # beta | rest ~ N(A^-1 X'y, sigma2 A^-1), A = X'X + diag(1/(tau2*lam2))
      A = XtX + np.diag(1.0 / (tau2 * lam2))
      try:
        L = np.linalg.cholesky(A)
      except np.linalg.LinAlgError:
        A += 1e-6 * np.eye(p)
        L = np.linalg.cholesky(A)
      mu = np.linalg.solve(A, Xty)
      z = rng.standard_normal(p)
      beta = mu + np.sqrt(sigma2) * np.linalg.solve(L.T, z)
      # sigma2 | rest ~ InvGamma
      resid = y - X @ beta
      shape = (n + p) / 2.0
      scale = resid @ resid / 2.0 + (beta ** 2 / (tau2 * lam2)).sum() / 2.0
      sigma2 = float(scale / max(rng.gamma(shape, 1.0), 1e-12))
      # lambda_j, nu_j
      lam2 = 1.0 / rng.gamma(1.0, 1.0 / (
          1.0 / nu + beta ** 2 / (2.0 * tau2 * sigma2)))
      nu = 1.0 / rng.gamma(1.0, 1.0 / (1.0 + 1.0 / lam2))
      # tau2, xi
      tau2 = float(1.0 / max(rng.gamma(
          (p + 1) / 2.0,
          1.0 / (1.0 / xi + (beta ** 2 / lam2).sum() / (2.0 * sigma2))),
          1e-12))
      xi = float(1.0 / max(rng.gamma(1.0, 1.0 / (1.0 + 1.0 / tau2)),
                           1e-12))
      if it >= self.n_iters // 2:
        betas.append(beta.copy())
    self.beta_samples = np.stack(betas)

  def sample_weights(self) -> np.ndarray:
    idx = self._rng.integers(0, len(self.beta_samples))
    return self.beta_samples[idx]


class BOCSDesigner(Designer):
  """Quadratic binary surrogate + simulated-annealing acquisition."""

  def __init__(self, problem: vz.ProblemStatement, *,
               order: int = 2, num_init_samples: int = 10,
               sa_iters: int = 500, seed: Optional[int] = None,
               acquisition_optimizer: str = 'sdp',
               lamda: float = 1e-4, sdp_repeats: int = 100,
               sdp_bm_iters: int = 300):
    if acquisition_optimizer not in ('sdp', 'sa'):
      raise ValueError("acquisition_optimizer must be 'sdp' or 'sa'")
    self._configs = _bool_configs(problem)
    self._n = len(self._configs)
    self._order = order
    self._num_init = num_init_samples
    self._sa_iters = sa_iters
    # SDP acquisition (the reference default, bocs.py:537-539).
    self._acquisition_optimizer = acquisition_optimizer
    self._lamda = lamda
    self._sdp_repeats = sdp_repeats
    self._sdp_bm_iters = sdp_bm_iters
    self._rng = np.random.default_rng(seed)
    self._xs: List[np.ndarray] = []
    self._ys: List[float] = []
    self._pairs: List[Tuple[int, int]] = list(
        itertools.combinations(range(self._n), 2)) if order >= 2 else []
    self._goal_sign = 1.0
    for mi in problem.metric_information:
      self._goal_sign = 1.0 if mi.goal.is_maximize else -1.0
      break

  def _features(self, xs: np.ndarray) -> np.ndarray:
    cols = [np.ones(len(xs))]
    cols.extend(xs[:, i] for i in range(self._n))
    cols.extend(xs[:, i] * xs[:, j] for i, j in self._pairs)
    return np.stack(cols, axis=1)

  def update(self, completed: CompletedTrials, all_active: ActiveTrials
             ) -> None:
    del all_active
    for t in completed.trials:
      if t.final_measurement is None or t.infeasible:
        continue
      metric = next(iter(t.final_measurement.metrics.values()), None)
      if metric is None:
        continue
      bits = np.array([1.0 if t.parameters.get_value(pc.name) == 'true'
                       else 0.0 for pc in self._configs])
      self._xs.append(bits)
      self._ys.append(self._goal_sign * metric.value)

  def _simulated_annealing(self, weights: np.ndarray) -> np.ndarray:
    x = self._rng.integers(0, 2, self._n).astype(np.float64)
    val = float((self._features(x[None, :]) @ weights)[0])
    best_x, best_val = x.copy(), val
    for it in range(self._sa_iters):
      temp = max(1e-3, 1.0 * (1 - it / self._sa_iters))
      flip = self._rng.integers(0, self._n)
      x2 = x.copy()
      x2[flip] = 1.0 - x2[flip]
      val2 = float((self._features(x2[None, :]) @ weights)[0])
      if val2 > val or self._rng.random() < np.exp((val2 - val) / temp):
        x, val = x2, val2
        if val > best_val:
          best_x, best_val = x.copy(), val
    return best_x

  def _sdp_rounding(self, weights: np.ndarray) -> np.ndarray:
    """SDP-relaxation acquisition (reference bocs.py:448-525).

    Minimizes x^T A x + b^T x over {0,1}^n via the +-1 homogenized SDP
    min tr(At X), X PSD, diag(X)=1, solved with a Burer-Monteiro
    low-rank factorization X = V^T V on the product of unit spheres
    (projected gradient descent — no external SDP solver needed
    offline; same relaxation cvxpy solves in the reference), followed
    by Goemans-Williamson random-hyperplane rounding.
    """
    n = self._n
    lamda = self._lamda
    # Internal convention maximizes; the SDP formulation minimizes.
    alpha = -weights
    b = alpha[1:n + 1] + lamda
    a = alpha[n + 1:]
    A = np.zeros((n, n))
    for k, (i, j) in enumerate(self._pairs):
      A[i, j] = a[k] / 2.0
      A[j, i] = a[k] / 2.0
    bt = b / 2.0 + A @ np.ones(n) / 2.0
    At = np.zeros((n + 1, n + 1))
    At[:n, :n] = A / 4.0
    At[:n, n] = bt / 2.0
    At[n, :n] = bt / 2.0
    At[n, n] = 2.0
    # Burer-Monteiro: V (k, n+1) with unit columns; grad tr(At V^T V)
    # = 2 V At; Riemannian step = project out the radial component.
    k_rank = max(2, int(np.ceil(np.sqrt(2.0 * (n + 1)))))
    V = self._rng.standard_normal((k_rank, n + 1))
    V /= np.linalg.norm(V, axis=0, keepdims=True)
    scale = max(np.abs(At).max(), 1e-12)
    lr = 0.2 / scale
    for it in range(self._sdp_bm_iters):
      grad = 2.0 * V @ At
      grad -= V * (V * grad).sum(axis=0, keepdims=True)
      V -= lr * grad
      V /= np.linalg.norm(V, axis=0, keepdims=True)
      if it == self._sdp_bm_iters // 2:
        lr *= 0.3
    # Random-hyperplane rounding; keep the best of y and -y per cut.
    best_x, best_obj = None, np.inf
    for _ in range(self._sdp_repeats):
      r = self._rng.standard_normal(k_rank)
      r /= max(np.linalg.norm(r), 1e-12)
      y = np.sign(r @ V)
      y[y == 0] = 1.0
      for cand in (y, -y):
        x01 = (cand[:n] + 1.0) / 2.0
        obj = float(x01 @ A @ x01 + b @ x01)
        if obj < best_obj:
          best_obj, best_x = obj, x01
    return best_x

  def suggest(self, count: Optional[int] = None
              ) -> Sequence[vz.TrialSuggestion]:
    count = count or 1
    out = []
    model = None
    if len(self._xs) >= self._num_init:
      model = BayesianHorseshoeLinearRegression(
          n_iters=100, seed=int(self._rng.integers(1 << 31)))
      model.fit(self._features(np.stack(self._xs)),
                np.asarray(self._ys, dtype=np.float64))
    for _ in range(count):
      if model is None:
        bits = self._rng.integers(0, 2, self._n)
      elif self._acquisition_optimizer == 'sdp' and self._pairs:
        bits = self._sdp_rounding(model.sample_weights())
      else:
        bits = self._simulated_annealing(model.sample_weights())
      params = {pc.name: ('true' if bits[i] > 0.5 else 'false')
                for i, pc in enumerate(self._configs)}
      out.append(vz.TrialSuggestion(params))
    return out
