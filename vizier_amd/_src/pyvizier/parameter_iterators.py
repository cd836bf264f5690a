"""Sequential builder for (conditional) parameter assignments.

Capability parity with vizier/_src/pyvizier/shared/
parameter_iterators.py (SequentialParameterBuilder :29): iterate the
search space one ParameterConfig at a time, choosing a value (or
skipping) at each step; chosen values unlock the matching conditional
child subspaces in DFS or BFS order.
"""

from __future__ import annotations

import copy
from typing import Generator, Iterator, Optional, Union

from vizier_amd._src.pyvizier.parameter_config import (
    ParameterConfig,
    ParameterValueTypes,
    SearchSpace,
)
from vizier_amd._src.pyvizier.trial import ParameterDict


class SequentialParameterBuilder(Iterator[ParameterConfig]):
  """Builds a ParameterDict by choosing one parameter value at a time.

  Usage:
    builder = SequentialParameterBuilder(search_space)
    for pc in builder:
      builder.choose_value(decide_value(pc))
    parameters = builder.parameters
  """

  def __init__(self, search_space: SearchSpace, *,
               traverse_order: str = 'dfs'):
    if traverse_order not in ('dfs', 'bfs'):
      raise ValueError(f'Bad traverse_order: {traverse_order}')
    self._parameters = ParameterDict()
    self._traverse_order = traverse_order
    self._gen = self._coroutine(search_space)
    self._next: Optional[ParameterConfig] = next(self._gen)
    self._stop: Optional[StopIteration] = None

  def _coroutine(
      self, search_space: SearchSpace
  ) -> Generator[ParameterConfig, Union[ParameterValueTypes, None], None]:
    space = copy.deepcopy(search_space)
    while space.parameters:
      config = space.parameters[0]
      value = yield config
      if value is None:          # skipped
        space.pop(config.name)
        continue
      subspace_map = config.subspaces_by_value
      child = subspace_map.get(value)
      subspace = copy.deepcopy(child) if child is not None \
          else SearchSpace()
      space.pop(config.name)
      self._parameters[config.name] = value
      if self._traverse_order == 'bfs':
        for p in subspace.parameters:
          space.add(copy.deepcopy(p))
      else:
        for p in space.parameters:
          subspace.add(copy.deepcopy(p))
        space = subspace

  def __next__(self) -> ParameterConfig:
    if self._stop is not None:
      raise self._stop
    assert self._next is not None
    return self._next

  def choose_value(self, value: ParameterValueTypes) -> None:
    try:
      self._next = self._gen.send(value)
    except StopIteration as e:
      self._stop = e

  def skip(self) -> None:
    try:
      self._next = self._gen.send(None)
    except StopIteration as e:
      self._stop = e

  @property
  def parameters(self) -> ParameterDict:
    return self._parameters
