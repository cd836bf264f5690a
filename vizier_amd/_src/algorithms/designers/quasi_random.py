"""Quasi-random (scrambled Halton) designer.

Capability parity with vizier/_src/algorithms/designers/quasi_random.py:32
(scipy qmc.Halton with serialized skip state; per-spec discretization).
Implements PartiallySerializableDesigner so the generator position
survives across suggest calls via study metadata.
"""

from __future__ import annotations

import math
from typing import Any, Dict, List, Optional

import numpy as np
from scipy.stats import qmc

from vizier_amd._src.pyvizier.base_study_config import ProblemStatement
from vizier_amd._src.pyvizier.common import Metadata
from vizier_amd._src.pyvizier.parameter_config import (
    ParameterConfig,
    ParameterType,
    ScaleType,
    SearchSpace,
)
from vizier_amd._src.pyvizier.trial import TrialSuggestion
from vizier_amd._src.algorithms.core.abstractions import (
    ActiveTrials,
    CompletedTrials,
    PartiallySerializableDesigner,
)

_NS = 'quasi_random'


def _unit_to_value(pc: ParameterConfig, u: float):
  """Maps u in [0,1) to a feasible value, honoring the scale type."""
  if pc.type == ParameterType.DOUBLE:
    lo, hi = pc.bounds
    if pc.scale_type == ScaleType.LOG and lo > 0:
      return float(math.exp(math.log(lo) + u * (math.log(hi) - math.log(lo))))
    if pc.scale_type == ScaleType.REVERSE_LOG and lo > 0:
      # Denser near the top of the range.
      span = hi - lo
      return float(hi + lo - math.exp(math.log(lo) + (1 - u) *
                                      (math.log(hi) - math.log(lo))))
    return float(lo + u * (hi - lo))
  if pc.type == ParameterType.INTEGER:
    lo, hi = pc.bounds
    return int(min(int(lo) + int(u * (int(hi) - int(lo) + 1)), int(hi)))
  values = pc.feasible_values
  return values[min(int(u * len(values)), len(values) - 1)]


class QuasiRandomDesigner(PartiallySerializableDesigner):
  """Scrambled Halton sequence over the (flattened) search space."""

  def __init__(self, search_space: SearchSpace, *, skip_points: int = 0,
               seed: Optional[int] = None):
    if search_space.is_conditional:
      raise ValueError(
          'QuasiRandomDesigner does not support conditional spaces.')
    if not search_space.parameters:
      raise ValueError('Empty search space.')
    self._search_space = search_space
    self._seed = 0 if seed is None else seed
    self._num_points = int(skip_points)
    self._halton = self._make_generator()

  @classmethod
  def from_problem(cls, problem: ProblemStatement,
                   seed: Optional[int] = None) -> 'QuasiRandomDesigner':
    return cls(problem.search_space, seed=seed)

  def _make_generator(self) -> qmc.Halton:
    gen = qmc.Halton(d=len(self._search_space.parameters), scramble=True,
                     seed=self._seed)
    if self._num_points:
      gen.fast_forward(self._num_points)
    return gen

  def update(self, completed: CompletedTrials, all_active: ActiveTrials
             ) -> None:
    del completed, all_active

  def suggest(self, count: Optional[int] = None) -> List[TrialSuggestion]:
    count = count or 1
    points = self._halton.random(count)
    self._num_points += count
    out = []
    for row in points:
      params: Dict[str, Any] = {}
      for pc, u in zip(self._search_space.parameters, row):
        params[pc.name] = _unit_to_value(pc, float(u))
      out.append(TrialSuggestion(params))
    return out

  # -- PartiallySerializableDesigner ---------------------------------------

  def dump(self) -> Metadata:
    md = Metadata()
    md.ns(_NS)['num_points'] = str(self._num_points)
    md.ns(_NS)['seed'] = str(self._seed)
    return md

  def load(self, metadata: Metadata) -> None:
    view = metadata.abs_ns((_NS,))
    self._num_points = int(view['num_points'])
    self._seed = int(view['seed'])
    self._halton = self._make_generator()
