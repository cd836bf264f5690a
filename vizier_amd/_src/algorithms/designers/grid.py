"""(Shuffled) grid-search designer.

Capability parity with vizier/_src/algorithms/designers/grid.py:36
(GridSearchDesigner with serialized index state and optional shuffling).
"""

from __future__ import annotations

import itertools
from typing import Any, Dict, List, Optional

import numpy as np

from vizier_amd._src.pyvizier.base_study_config import ProblemStatement
from vizier_amd._src.pyvizier.common import Metadata
from vizier_amd._src.pyvizier.parameter_config import (
    ParameterType,
    SearchSpace,
)
from vizier_amd._src.pyvizier.trial import TrialSuggestion
from vizier_amd._src.algorithms.core.abstractions import (
    ActiveTrials,
    CompletedTrials,
    PartiallySerializableDesigner,
)

_NS = 'grid'


class GridSearchDesigner(PartiallySerializableDesigner):
  """Iterates the Cartesian product of per-parameter grids."""

  def __init__(self, search_space: SearchSpace, *, shuffle_seed:
               Optional[int] = None, double_grid_resolution: int = 10):
    if search_space.is_conditional:
      raise ValueError('GridSearchDesigner does not support conditional '
                       'spaces.')
    self._search_space = search_space
    self._shuffle_seed = shuffle_seed
    self._resolution = double_grid_resolution
    self._index = 0
    self._grid_values: List[List[Any]] = []
    for pc in search_space.parameters:
      if pc.type == ParameterType.DOUBLE:
        lo, hi = pc.bounds
        vals = list(np.linspace(lo, hi, double_grid_resolution))
      else:
        vals = list(pc.feasible_values)
      self._grid_values.append(vals)
    self._total = 1
    for vals in self._grid_values:
      self._total *= len(vals)
    self._order = None
    if shuffle_seed is not None:
      rng = np.random.default_rng(shuffle_seed)
      self._order = rng.permutation(self._total)

  @classmethod
  def from_problem(cls, problem: ProblemStatement,
                   shuffle_seed: Optional[int] = None) -> 'GridSearchDesigner':
    return cls(problem.search_space, shuffle_seed=shuffle_seed)

  def _point(self, flat_index: int) -> Dict[str, Any]:
    flat_index %= self._total
    if self._order is not None:
      flat_index = int(self._order[flat_index])
    params = {}
    for pc, vals in zip(reversed(self._search_space.parameters),
                        reversed(self._grid_values)):
      flat_index, rem = divmod(flat_index, len(vals))
      params[pc.name] = vals[rem]
    return dict(reversed(list(params.items())))

  def update(self, completed: CompletedTrials, all_active: ActiveTrials
             ) -> None:
    del completed, all_active

  def suggest(self, count: Optional[int] = None) -> List[TrialSuggestion]:
    count = count or 1
    out = [TrialSuggestion(self._point(self._index + i))
           for i in range(count)]
    self._index += count
    return out

  def dump(self) -> Metadata:
    md = Metadata()
    md.ns(_NS)['index'] = str(self._index)
    return md

  def load(self, metadata: Metadata) -> None:
    self._index = int(metadata.abs_ns((_NS,))['index'])
