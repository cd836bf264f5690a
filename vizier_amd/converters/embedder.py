"""Problem statement and trials scaler ("embedder").

Capability parity with vizier/pyvizier/converters/embedder.py
(ProblemAndTrialsScaler :44): produces a scaled copy of a
ProblemStatement and maps trials between the original and the scaled
("embedded") search space:

- DOUBLE / INTEGER parameters -> FLOAT in [0, 1] (converter scaling,
  honoring LINEAR / LOG / REVERSE_LOG scale types),
- DISCRETE parameters -> DISCRETE with scaled feasible values,
- CATEGORICAL parameters unchanged.
"""

from __future__ import annotations

import copy
from typing import List, Sequence, TypeVar, Union

from vizier_amd import pyvizier as vz
from vizier_amd.converters import core as converters_core

_T = TypeVar('_T', vz.Trial, vz.TrialSuggestion)


def _evolve(trial: _T, parameters: vz.ParameterDict) -> _T:
  out = copy.deepcopy(trial)
  out.parameters = parameters
  return out


class ProblemAndTrialsScaler:
  """Scales a problem to the unit cube and maps trials to/from it."""

  def __init__(self, problem: vz.ProblemStatement):
    self._configs = {p.name: p for p in problem.search_space.parameters}
    space = vz.SearchSpace()
    for param in problem.search_space.parameters:
      if param.type in (vz.ParameterType.DOUBLE,
                        vz.ParameterType.INTEGER):
        space.root.add_float_param(param.name, 0.0, 1.0)
      elif param.type == vz.ParameterType.DISCRETE:
        space.root.add_discrete_param(
            param.name,
            feasible_values=[
                converters_core._scale(param, float(v))
                for v in param.feasible_values])
      elif param.type == vz.ParameterType.CATEGORICAL:
        space.root.add_categorical_param(
            param.name, feasible_values=list(param.feasible_values))
      else:
        raise ValueError(f'Unsupported parameter type {param.type}')
    self._embedded = copy.deepcopy(problem)
    self._embedded.search_space = space

  @property
  def problem_statement(self) -> vz.ProblemStatement:
    return self._embedded

  def map(self, trials: Sequence[_T]) -> List[_T]:
    """Original space -> embedded (scaled) space."""
    out = []
    for trial in trials:
      params = vz.ParameterDict()
      for name, pv in trial.parameters.items():
        cfg = self._configs.get(name)
        if cfg is None or cfg.type == vz.ParameterType.CATEGORICAL:
          params[name] = pv.value
        else:
          params[name] = converters_core._scale(cfg, float(pv.value))
      out.append(_evolve(trial, params))
    return out

  def unmap(self, trials: Sequence[_T]) -> List[_T]:
    """Embedded (scaled) space -> original space (with rounding for
    INTEGER/DISCRETE parameters)."""
    out = []
    for trial in trials:
      params = vz.ParameterDict()
      for name, pv in trial.parameters.items():
        cfg = self._configs.get(name)
        if cfg is None or cfg.type == vz.ParameterType.CATEGORICAL:
          params[name] = pv.value
        else:
          value = converters_core._unscale(cfg, float(pv.value))
          if cfg.type in (vz.ParameterType.INTEGER,
                          vz.ParameterType.DISCRETE):
            value = cfg.round_to_feasible(value)
          params[name] = value
      out.append(_evolve(trial, params))
    return out
