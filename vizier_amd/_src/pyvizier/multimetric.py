"""Multi-objective utilities: Pareto dominance, hypervolume, safety.

Capability parity with vizier/_src/pyvizier/multimetric/
(pareto_optimal.py:87,121; hypervolume.py:25,68; safety.py:24).
All points use the maximize-everything convention (goals pre-flipped).
The heavy-sweep variants of these kernels live on the GPU in
vizier_amd/_src/gp (device dominance + hypervolume-scalarization); these
NumPy versions serve the service's `ListOptimalTrials` and analysis paths.
"""

from __future__ import annotations

import math
from typing import Iterable, List, Optional

import numpy as np

from vizier_amd._src.pyvizier.base_study_config import (
    MetricsConfig,
    MetricType,
)
from vizier_amd._src.pyvizier.trial import Measurement, Trial


def is_pareto_optimal_against(points: np.ndarray, against: np.ndarray, *,
                              strict: bool = False,
                              block: int = 2048) -> np.ndarray:
  """For each point, True iff no row of `against` dominates it.

  A point q is dominated by p iff p >= q componentwise with p != q
  (and `strict=True` additionally treats exact duplicates as optimal).
  """
  points = np.asarray(points, dtype=np.float64)
  against = np.asarray(against, dtype=np.float64)
  if len(points) == 0:
    return np.zeros(0, dtype=bool)
  if len(against) == 0:
    return np.ones(len(points), dtype=bool)
  out = np.ones(len(points), dtype=bool)
  for start in range(0, len(points), block):
    chunk = points[start:start + block]  # (b, m)
    geq = np.all(against[None, :, :] >= chunk[:, None, :], axis=2)
    if strict:
      # Only strictly-better rows dominate: >= everywhere and > somewhere.
      # Exact duplicates therefore stay optimal.
      gt = np.any(against[None, :, :] > chunk[:, None, :], axis=2)
      dominated = np.any(geq & gt, axis=1)
    else:
      # Weak dominance: an equal (or better) row in `against` dominates.
      dominated = np.any(geq, axis=1)
    out[start:start + block] = ~dominated
  return out


def is_pareto_optimal(points: np.ndarray, *, block: int = 2048) -> np.ndarray:
  """True for each point on the Pareto frontier (duplicates kept optimal)."""
  points = np.asarray(points, dtype=np.float64)
  if len(points) == 0:
    return np.zeros(0, dtype=bool)
  return is_pareto_optimal_against(points, points, strict=True, block=block)


class NaiveParetoOptimalAlgorithm:
  """API-parity wrapper over the vectorized implementation."""

  def is_pareto_optimal(self, points: np.ndarray) -> np.ndarray:
    return is_pareto_optimal(points)

  def is_pareto_optimal_against(self, points: np.ndarray,
                                against: np.ndarray, *,
                                strict: bool = False) -> np.ndarray:
    return is_pareto_optimal_against(points, against, strict=strict)


class FastParetoOptimalAlgorithm(NaiveParetoOptimalAlgorithm):
  """Divide-and-conquer frontier computation for large point sets."""

  def __init__(self, base_algorithm: Optional[
      NaiveParetoOptimalAlgorithm] = None, *,
               recursive_threshold: int = 10000):
    self._base = base_algorithm or NaiveParetoOptimalAlgorithm()
    self._threshold = recursive_threshold

  def is_pareto_optimal(self, points: np.ndarray) -> np.ndarray:
    points = np.asarray(points, dtype=np.float64)
    if len(points) <= self._threshold:
      return self._base.is_pareto_optimal(points)
    # Split by the first objective; the top half cannot be dominated by the
    # bottom half except via equal first coordinate, handled by the
    # cross-check below.
    order = np.argsort(-points[:, 0], kind='stable')
    half = len(points) // 2
    top, bottom = order[:half], order[half:]
    res = np.zeros(len(points), dtype=bool)
    res[top] = self.is_pareto_optimal(points[top])
    bottom_opt = self.is_pareto_optimal(points[bottom])
    # Bottom survivors must also be optimal against top survivors.
    surv = bottom[bottom_opt]
    if len(surv):
      against = points[top[res[top]]]
      res[surv] = self._base.is_pareto_optimal_against(
          points[surv], against, strict=True)
    return res


class ParetoFrontier:
  """Randomized-direction hypervolume approximation.

  Implements the scalarization estimator of arXiv:2006.04655 (Lemma 5),
  matching the reference's ParetoFrontier (hypervolume.py:68): sample
  directions |N(0,I)| normalized to the unit sphere's positive orthant;
  the dominated hypervolume is the mean over directions of
  (max over points of min over dims (y/lambda))^m scaled by the volume of
  the positive-orthant unit ball.
  """

  def __init__(self, points: np.ndarray, origin: np.ndarray,
               num_vectors: int = 10000, seed: Optional[int] = None):
    self._points = np.asarray(points, dtype=np.float64)
    origin = np.asarray(origin, dtype=np.float64).reshape(-1)
    if self._points.shape[1] != len(origin):
      raise ValueError(
          f'Dimension mismatch: points {self._points.shape} vs origin '
          f'{origin.shape}')
    self._origin = origin
    rng = np.random.default_rng(seed)
    vecs = np.abs(rng.standard_normal((num_vectors, len(origin))))
    self._vectors = vecs / np.linalg.norm(vecs, axis=1, keepdims=True)

  def hypervolume(self, additional_points: Optional[np.ndarray] = None,
                  is_cumulative: bool = False,
                  num_shards: int = 10) -> np.ndarray:
    points = self._points
    if additional_points is not None:
      additional_points = np.asarray(additional_points, dtype=np.float64)
      if points.shape[1] != additional_points.shape[1]:
        raise ValueError('Dimension mismatch for additional points.')
      points = np.concatenate([points, additional_points], axis=0)
    points = points - self._origin
    # Points not strictly dominating the origin contribute nothing.
    points = np.where(np.all(points > 0, axis=1, keepdims=True), points, 0.0)
    if points.size == 0:
      return np.asarray([0.0])

    m = points.shape[1]
    orthant_ball_volume = (math.pi ** (m / 2) / math.gamma(m / 2 + 1)
                           / 2 ** m)
    idx = np.linspace(0, len(self._vectors), num_shards + 1).astype(int)
    acc = 0.0
    for begin, end in zip(idx[:-1], idx[1:]):
      vectors = self._vectors[begin:end]
      # ratios: (V, P) = min over dims of point/vector.
      ratios = np.min(points[None, :, :] / vectors[:, None, :], axis=2)
      prefix_max = np.maximum.accumulate(ratios, axis=1)
      acc += np.mean(prefix_max ** m, axis=0) * orthant_ball_volume
    acc = np.maximum.accumulate(acc / num_shards)
    return np.array(acc) if is_cumulative else np.array(np.max(acc))


class SafetyChecker:
  """Evaluates safety-metric constraints on measurements/trials.

  Parity with vizier/_src/pyvizier/multimetric/safety.py:24. A measurement
  is safe iff every present safety metric respects its threshold (missing
  metrics are assumed safe).
  """

  def __init__(self, metrics_config: MetricsConfig):
    self._safety_metrics = list(metrics_config.of_type(MetricType.SAFETY))

  def are_trials_safe(self, trials: Iterable[Trial]) -> List[bool]:
    return self.are_measurements_safe(
        t.final_measurement if t.final_measurement else Measurement()
        for t in trials)

  def are_measurements_safe(self, measurements: Iterable[Measurement]
                            ) -> List[bool]:
    out = []
    for m in measurements:
      safe = True
      for cfg in self._safety_metrics:
        metric = m.metrics.get(cfg.name)
        if metric is None:
          continue
        if cfg.goal.is_maximize and metric.value < cfg.safety_threshold:
          safe = False
        elif cfg.goal.is_minimize and metric.value > cfg.safety_threshold:
          safe = False
      out.append(safe)
    return out
