"""Uniform random designer + sampling helpers.

Capability parity with vizier/_src/algorithms/designers/random.py:27 and
random/random_sample.py:28-121 (sampling helpers over search spaces,
including conditional children).
"""

from __future__ import annotations

import math
from typing import Any, Dict, Optional, Sequence

import numpy as np

from vizier_amd._src.pyvizier.base_study_config import ProblemStatement
from vizier_amd._src.pyvizier.parameter_config import (
    ParameterConfig,
    ParameterType,
    ScaleType,
)
from vizier_amd._src.pyvizier.trial import TrialSuggestion
from vizier_amd._src.algorithms.core.abstractions import (
    ActiveTrials,
    CompletedTrials,
    Designer,
)


def sample_value(rng: np.random.Generator, pc: ParameterConfig):
  """Samples one feasible value of `pc` uniformly (scale-aware for DOUBLE)."""
  if pc.type == ParameterType.DOUBLE:
    lo, hi = pc.bounds
    if pc.scale_type == ScaleType.LOG and lo > 0:
      return float(math.exp(rng.uniform(math.log(lo), math.log(hi))))
    return float(rng.uniform(lo, hi))
  if pc.type == ParameterType.INTEGER:
    lo, hi = pc.bounds
    return int(rng.integers(int(lo), int(hi) + 1))
  # DISCRETE / CATEGORICAL
  values = pc.feasible_values
  return values[int(rng.integers(0, len(values)))]


def sample_parameters(rng: np.random.Generator, configs: Sequence[
    ParameterConfig], out: Optional[Dict[str, Any]] = None) -> Dict[str, Any]:
  """Samples a full (possibly conditional) assignment."""
  if out is None:
    out = {}
  for pc in configs:
    value = sample_value(rng, pc)
    out[pc.name] = value
    sub = pc.subspaces_by_value.get(value)
    if sub is not None:
      sample_parameters(rng, sub.parameters, out)
  return out


class RandomDesigner(Designer):
  """Samples suggestions uniformly at random from the search space."""

  def __init__(self, search_space, *, seed: Optional[int] = None):
    if search_space is None or not search_space.parameters:
      raise ValueError('RandomDesigner requires a non-empty search space.')
    self._search_space = search_space
    self._rng = np.random.default_rng(seed)

  @classmethod
  def from_problem(cls, problem: ProblemStatement,
                   seed: Optional[int] = None) -> 'RandomDesigner':
    return cls(problem.search_space, seed=seed)

  def update(self, completed: CompletedTrials, all_active: ActiveTrials
             ) -> None:
    del completed, all_active  # Stateless.

  def suggest(self, count: Optional[int] = None):
    count = count or 1
    return [TrialSuggestion(sample_parameters(
        self._rng, self._search_space.parameters)) for _ in range(count)]
