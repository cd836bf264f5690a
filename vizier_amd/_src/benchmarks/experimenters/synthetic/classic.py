"""Classic synthetic experimenters: Branin, Hartmann, DH, multi-arm.

Capability parity with vizier/_src/benchmarks/experimenters/synthetic/
branin.py (:51), hartmann.py (:34), deb.py (DHExperimenter :35,
DH1-DH4), multiarm.py (Bernoulli/FixedMultiArmExperimenter).
"""

from __future__ import annotations

import copy
import math
from typing import Callable, Mapping, Optional, Sequence, Tuple

import numpy as np

from vizier_amd import pyvizier as vz
from vizier_amd._src.benchmarks.experimenters.experimenter import (
    Experimenter,
)
from vizier_amd._src.benchmarks.experimenters.numpy_experimenter import (
    NumpyExperimenter,
)


def branin(x: np.ndarray) -> float:
  """The 2D Branin function (https://www.sfu.ca/~ssurjano/branin.html)."""
  a, r, s = 1.0, 6.0, 10.0
  b = 5.1 / (4 * math.pi ** 2)
  c = 5 / math.pi
  t = 1 / (8 * math.pi)
  x1, x2 = x[..., 0], x[..., 1]
  return float(a * (x2 - b * x1 ** 2 + c * x1 - r) ** 2 +
               s * (1 - t) * np.cos(x1) + s)


class Branin2DExperimenter(Experimenter):
  """2D minimization on [-5,10] x [0,15]; min value 0.397887."""

  def __init__(self):
    self._impl = NumpyExperimenter(branin, self.problem_statement())

  def evaluate(self, suggestions: Sequence[vz.Trial]) -> None:
    self._impl.evaluate(suggestions)

  def problem_statement(self) -> vz.ProblemStatement:
    problem = vz.ProblemStatement()
    problem.search_space.root.add_float_param('x1', -5, 10)
    problem.search_space.root.add_float_param('x2', 0, 15)
    problem.metric_information.append(vz.MetricInformation(
        name='value', goal=vz.ObjectiveMetricGoal.MINIMIZE))
    return problem


class HartmannExperimenter(Experimenter):
  """General Hartmann minimization family on the unit cube."""

  def __init__(self, alpha: np.ndarray, A: np.ndarray, P: np.ndarray):
    self._alpha = np.asarray(alpha, dtype=float)
    self._A = np.asarray(A, dtype=float)
    self._P = np.asarray(P, dtype=float)
    self._dimension = self._A.shape[-1]
    self._impl = NumpyExperimenter(self._fn, self.problem_statement())

  def _fn(self, x: np.ndarray) -> float:
    return float(-self._alpha @ np.exp(
        -np.sum(self._A * (x - self._P) ** 2, axis=1)))

  def evaluate(self, suggestions: Sequence[vz.Trial]) -> None:
    self._impl.evaluate(suggestions)

  def problem_statement(self) -> vz.ProblemStatement:
    problem = vz.ProblemStatement()
    for i in range(1, self._dimension + 1):
      problem.search_space.root.add_float_param(f'x{i}', 0, 1)
    problem.metric_information.append(vz.MetricInformation(
        name='value', goal=vz.ObjectiveMetricGoal.MINIMIZE))
    return problem

  @classmethod
  def from_3d(cls) -> 'HartmannExperimenter':
    return cls(
        alpha=np.array([1.0, 1.2, 3.0, 3.2]),
        A=np.array([[3, 10, 30], [0.1, 10, 35], [3, 10, 30],
                    [0.1, 10, 35]]),
        P=1e-4 * np.array([[3689, 1170, 2673], [4699, 4387, 7470],
                           [1091, 8732, 5547], [381, 5743, 8828]]))

  @classmethod
  def from_6d(cls) -> 'HartmannExperimenter':
    return cls(
        alpha=np.array([1.0, 1.2, 3.0, 3.2]),
        A=np.array([[10, 3, 17, 3.5, 1.7, 8],
                    [0.05, 10, 17, 0.1, 8, 14],
                    [3, 3.5, 1.7, 10, 17, 8],
                    [17, 8, 0.05, 10, 0.1, 14]]),
        P=1e-4 * np.array([
            [1312, 1696, 5569, 124, 8283, 5886],
            [2329, 4135, 8307, 3736, 1004, 9991],
            [2348, 1451, 3522, 2883, 3047, 6650],
            [4047, 8828, 8732, 5743, 1091, 381]]))


class DHExperimenter(Experimenter):
  """Deb-Gupta robust multi-objective family (deb.py:35).

  f0(x) = x0; f1(x) = h + g*s (DH1/DH2) or h*(g+s) (DH3/DH4), both
  MINIMIZE.
  """

  def __init__(self, h_fn: Callable[[np.ndarray], float],
               g_fn: Callable[[np.ndarray], float],
               s_fn: Callable[[np.ndarray], float],
               f1_fn: Callable[[float, float, float], float],
               bounds: Sequence[Tuple[float, float]]):
    self._h_fn, self._g_fn, self._s_fn = h_fn, g_fn, s_fn
    self._f1_fn = f1_fn
    self._bounds = list(bounds)

  def problem_statement(self) -> vz.ProblemStatement:
    problem = vz.ProblemStatement()
    problem.metric_information.append(vz.MetricInformation(
        name='f0', goal=vz.ObjectiveMetricGoal.MINIMIZE))
    problem.metric_information.append(vz.MetricInformation(
        name='f1', goal=vz.ObjectiveMetricGoal.MINIMIZE))
    for i, (lo, hi) in enumerate(self._bounds):
      problem.search_space.root.add_float_param(f'x{i}', lo, hi)
    return problem

  def evaluate(self, suggestions: Sequence[vz.Trial]) -> None:
    for trial in suggestions:
      x = np.array([trial.parameters.get_value(f'x{i}')
                    for i in range(len(self._bounds))])
      f0 = float(x[0])
      f1 = float(self._f1_fn(self._h_fn(x), self._g_fn(x), self._s_fn(x)))
      trial.complete(vz.Measurement(metrics={'f0': f0, 'f1': f1}))

  @classmethod
  def DH1(cls, num_dimensions: int) -> 'DHExperimenter':
    if num_dimensions < 2:
      raise ValueError(f'num_dimensions must be >= 2: {num_dimensions}')
    return cls(
        h_fn=lambda x: 1 - x[0] ** 2,
        g_fn=lambda x: np.sum(10 + x[1:] ** 2 -
                              10 * np.cos(4 * np.pi * x[1:])),
        s_fn=lambda x: 1 / (0.2 + x[0]) + x[0] ** 2,
        f1_fn=lambda h, g, s: h + g * s,
        bounds=[(0, 1)] + [(-1, 1)] * (num_dimensions - 1))

  @classmethod
  def DH2(cls, num_dimensions: int) -> 'DHExperimenter':
    if num_dimensions < 2:
      raise ValueError(f'num_dimensions must be >= 2: {num_dimensions}')
    return cls(
        h_fn=lambda x: 1 - x[0] ** 2,
        g_fn=lambda x: np.sum(10 + x[1:] ** 2 -
                              10 * np.cos(4 * np.pi * x[1:])),
        s_fn=lambda x: 1 / (0.2 + x[0]) + 10.0 * x[0] ** 2,
        f1_fn=lambda h, g, s: h + g * s,
        bounds=[(0, 1)] + [(-1, 1)] * (num_dimensions - 1))

  @classmethod
  def DH3(cls, num_dimensions: int) -> 'DHExperimenter':
    if num_dimensions < 3:
      raise ValueError(f'num_dimensions must be >= 3: {num_dimensions}')
    return cls(
        h_fn=lambda x: (2 - 0.8 * np.exp(-((x[1] - 0.35) / 0.25) ** 2) -
                        np.exp(-((x[1] - 0.85) / 0.03) ** 2)),
        g_fn=lambda x: 50 * np.sum(x[2:] ** 2),
        s_fn=lambda x: 1 - np.sqrt(x[0]),
        f1_fn=lambda h, g, s: h * (g + s),
        bounds=[(0, 1), (0, 1)] + [(-1, 1)] * (num_dimensions - 2))

  @classmethod
  def DH4(cls, num_dimensions: int) -> 'DHExperimenter':
    if num_dimensions < 3:
      raise ValueError(f'num_dimensions must be >= 3: {num_dimensions}')
    return cls(
        h_fn=lambda x: (2 - x[0] -
                        0.8 * np.exp(-((np.sum(x[:2]) - 0.35) / 0.25) ** 2)
                        - np.exp(-((np.sum(x[:2]) - 0.85) / 0.03) ** 2)),
        g_fn=lambda x: 50 * np.sum(x[2:] ** 2),
        s_fn=lambda x: 1 - np.sqrt(x[0]),
        f1_fn=lambda h, g, s: h * (g + s),
        bounds=[(0, 1), (0, 1)] + [(-1, 1)] * (num_dimensions - 2))


def _multiarm_problem(arms: Sequence[str]) -> vz.ProblemStatement:
  problem = vz.ProblemStatement()
  problem.metric_information.append(vz.MetricInformation(
      name='reward', goal=vz.ObjectiveMetricGoal.MAXIMIZE))
  problem.search_space.root.add_categorical_param(
      'arm', feasible_values=list(arms))
  return problem


class BernoulliMultiArmExperimenter(Experimenter):
  """Each arm pays 0/1 reward with a fixed success probability."""

  def __init__(self, arms_to_probs: Mapping[str, float],
               seed: Optional[int] = None):
    self._arms_to_probs = dict(arms_to_probs)
    self._rng = np.random.RandomState(seed)

  def problem_statement(self) -> vz.ProblemStatement:
    return _multiarm_problem(self._arms_to_probs.keys())

  def evaluate(self, suggestions: Sequence[vz.Trial]) -> None:
    for trial in suggestions:
      p = self._arms_to_probs[str(trial.parameters['arm'].value)]
      reward = float(self._rng.choice([0, 1], p=[1 - p, p]))
      trial.final_measurement = vz.Measurement(metrics={'reward': reward})


class FixedMultiArmExperimenter(Experimenter):
  """Deterministic per-arm rewards."""

  def __init__(self, arms_to_rewards: Mapping[str, float]):
    self._arms_to_rewards = dict(arms_to_rewards)

  def problem_statement(self) -> vz.ProblemStatement:
    return _multiarm_problem(self._arms_to_rewards.keys())

  def evaluate(self, suggestions: Sequence[vz.Trial]) -> None:
    for trial in suggestions:
      reward = self._arms_to_rewards[str(trial.parameters['arm'].value)]
      trial.final_measurement = vz.Measurement(
          metrics={'reward': float(reward)})
