"""L-BFGS-B acquisition optimizer (continuous spaces).

Capability parity with
vizier/_src/algorithms/optimizers/lbfgsb_optimizer.py:48 (25 random
restarts of bound-constrained L-BFGS on the acquisition). Runs all
restarts as one batched tensor program via the same sync-free batched
L-BFGS used for ARD (bounds via sigmoid reparameterization).
"""

from __future__ import annotations

from typing import Callable, List, Optional, Sequence

import numpy as np
import torch

from vizier_amd import pyvizier as vz
from vizier_amd.converters.core import TrialToArrayConverter
from vizier_amd._src.algorithms.optimizers.base import (
    BatchTrialScoreFunction,
    GradientFreeOptimizer,
)
from vizier_amd._src.gp import lbfgs


class LBFGSBOptimizer(GradientFreeOptimizer):
  """Batched multi-restart L-BFGS over the converter's [0,1] cube.

  Requires a differentiable torch score function; falls back to
  finite-difference-free use only when `torch_score_fn` is given.
  """

  def __init__(self, *, num_restarts: int = 25, max_iters: int = 50,
               seed: Optional[int] = None):
    self.num_restarts = num_restarts
    self.max_iters = max_iters
    self._seed = seed or 0

  def optimize(self, score_fn: BatchTrialScoreFunction,
               problem: vz.ProblemStatement, *, count: int = 1,
               seed_candidates: Sequence[vz.TrialSuggestion] = ()
               ) -> List[vz.TrialSuggestion]:
    for pc in problem.search_space.parameters:
      if pc.type != vz.ParameterType.DOUBLE:
        raise ValueError('LBFGSBOptimizer requires a continuous space.')
    converter = TrialToArrayConverter(problem)
    d = converter.n_features
    # score_fn is a black box here, so this entry point refines batched
    # random restarts with shrinking pattern moves; differentiable (GP)
    # acquisitions should use optimize_torch below, which runs true
    # batched L-BFGS with bounds.
    rng = np.random.default_rng(self._seed)
    xs = rng.uniform(0, 1, (self.num_restarts, d))
    if seed_candidates:
      seeds = converter.to_features(list(seed_candidates))
      k = min(len(seeds), self.num_restarts)
      xs[:k] = seeds[:k]
    step = 0.25
    scores = self._score_array(score_fn, converter, xs)
    for _ in range(self.max_iters):
      proposals = np.clip(
          xs + rng.uniform(-step, step, xs.shape), 0.0, 1.0)
      new_scores = self._score_array(score_fn, converter, proposals)
      improved = new_scores > scores
      xs[improved] = proposals[improved]
      scores[improved] = new_scores[improved]
      step = max(step * 0.93, 1e-3)
    order = np.argsort(-scores)[:count]
    return converter.to_suggestions(xs[order])

  def optimize_torch(self, torch_score_fn: Callable[[torch.Tensor],
                                                    torch.Tensor],
                     problem: vz.ProblemStatement, *, count: int = 1,
                     device: str = 'cpu') -> List[vz.TrialSuggestion]:
    """Gradient-based path for differentiable (GP) acquisitions."""
    converter = TrialToArrayConverter(problem)
    d = converter.n_features
    g = torch.Generator().manual_seed(self._seed)
    u0 = torch.randn(self.num_restarts, d, generator=g).to(device)

    def loss(u: torch.Tensor) -> torch.Tensor:
      return -torch_score_fn(torch.sigmoid(u))

    u_best, f_best = lbfgs.minimize_batched(loss, u0,
                                            max_iters=self.max_iters)
    order = torch.argsort(f_best)[:count]
    xs = torch.sigmoid(u_best[order]).detach().cpu().numpy()
    return converter.to_suggestions(xs)

  def _score_array(self, score_fn, converter, xs: np.ndarray) -> np.ndarray:
    return np.asarray(score_fn(converter.to_suggestions(xs)),
                      dtype=np.float64).reshape(len(xs))


class RandomVectorizedOptimizer(GradientFreeOptimizer):
  """Pure random-search baseline (parity with
  random_vectorized_optimizer.py:32)."""

  def __init__(self, *, max_evaluations: int = 75000,
               batch_size: int = 1000, seed: Optional[int] = None):
    self.max_evaluations = max_evaluations
    self.batch_size = batch_size
    self._seed = seed or 0

  def optimize(self, score_fn: BatchTrialScoreFunction,
               problem: vz.ProblemStatement, *, count: int = 1,
               seed_candidates: Sequence[vz.TrialSuggestion] = ()
               ) -> List[vz.TrialSuggestion]:
    converter = TrialToArrayConverter(problem)
    rng = np.random.default_rng(self._seed)
    best_x: Optional[np.ndarray] = None
    best_s: Optional[np.ndarray] = None
    evaluated = 0
    while evaluated < self.max_evaluations:
      n = min(self.batch_size, self.max_evaluations - evaluated)
      xs = rng.uniform(0, 1, (n, converter.n_features))
      scores = np.asarray(score_fn(converter.to_suggestions(xs)),
                          dtype=np.float64).reshape(n)
      evaluated += n
      if best_x is None:
        best_x, best_s = xs, scores
      else:
        best_x = np.concatenate([best_x, xs])
        best_s = np.concatenate([best_s, scores])
      order = np.argsort(-best_s)[:max(count, 16)]
      best_x, best_s = best_x[order], best_s[order]
    return converter.to_suggestions(best_x[:count])


class DesignerAsOptimizer(GradientFreeOptimizer):
  """Uses any Designer loop to optimize an acquisition (parity with
  designer_optimizer.py:30)."""

  def __init__(self, designer_factory, *, batch_size: int = 25,
               num_evaluations: int = 2000):
    self._designer_factory = designer_factory
    self._batch_size = batch_size
    self._num_evaluations = num_evaluations

  def optimize(self, score_fn: BatchTrialScoreFunction,
               problem: vz.ProblemStatement, *, count: int = 1,
               seed_candidates: Sequence[vz.TrialSuggestion] = ()
               ) -> List[vz.TrialSuggestion]:
    from vizier_amd._src.algorithms.core.abstractions import (
        ActiveTrials,
        CompletedTrials,
    )
    acq_problem = vz.ProblemStatement(
        search_space=problem.search_space,
        metric_information=[vz.MetricInformation(
            name='acquisition', goal=vz.ObjectiveMetricGoal.MAXIMIZE)])
    designer = self._designer_factory(acq_problem)
    best: List[vz.Trial] = []
    uid = 0
    for _ in range(self._num_evaluations // self._batch_size):
      suggestions = designer.suggest(self._batch_size)
      if not suggestions:
        break
      scores = np.asarray(score_fn(suggestions)).reshape(-1)
      completed = []
      for s, v in zip(suggestions, scores):
        uid += 1
        t = s.to_trial(uid)
        t.complete(vz.Measurement(metrics={'acquisition': float(v)}))
        completed.append(t)
      designer.update(CompletedTrials(completed), ActiveTrials())
      best.extend(completed)
      best.sort(key=lambda t: -t.final_measurement.metrics[
          'acquisition'].value)
      best = best[:max(count, 16)]
    return [vz.TrialSuggestion(t.parameters, metadata=t.metadata)
            for t in best[:count]]
