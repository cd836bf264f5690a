"""Experimenter over a NumPy function (parity with
vizier/_src/benchmarks/experimenters/numpy_experimenter.py:40)."""

from __future__ import annotations

import copy
from typing import Callable, Sequence

import numpy as np

from vizier_amd import pyvizier as vz
from vizier_amd._src.benchmarks.experimenters.experimenter import (
    Experimenter,
)


class NumpyExperimenter(Experimenter):
  """Evaluates f(x) over the problem's (ordered) double parameters."""

  def __init__(self, impl: Callable[[np.ndarray], float],
               problem_statement: vz.ProblemStatement):
    self._impl = impl
    self._problem = problem_statement
    self._metric_name = problem_statement.metric_information.item().name
    self._param_names = [
        pc.name for pc in problem_statement.search_space.parameters]

  def evaluate(self, suggestions: Sequence[vz.Trial]) -> None:
    for trial in suggestions:
      x = np.array([trial.parameters.get_value(name)
                    for name in self._param_names], dtype=np.float64)
      value = float(self._impl(x))
      trial.complete(vz.Measurement(metrics={self._metric_name: value}))

  def problem_statement(self) -> vz.ProblemStatement:
    return copy.deepcopy(self._problem)

  def __repr__(self) -> str:
    return f'NumpyExperimenter({getattr(self._impl, "__name__", "fn")})'
