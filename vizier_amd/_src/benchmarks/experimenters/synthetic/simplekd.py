"""Simple-KD testbed: a mixed-type problem with a known optimum.

Capability parity with the reference's simplekd experimenter
(vizier/_src/benchmarks/experimenters/synthetic/simplekd.py), used by
designer convergence tests: one categorical, one discrete, one integer
and one continuous parameter; the objective rewards hitting the `best
category` and extremal settings of the numeric parameters.
"""

from __future__ import annotations

import copy
from typing import Sequence

import numpy as np

from vizier_amd import pyvizier as vz
from vizier_amd._src.benchmarks.experimenters.experimenter import (
    Experimenter,
)

_CATEGORIES = ('corner', 'center', 'mixed')


class SimpleKDExperimenter(Experimenter):
  """Mixed-type single-objective testbed with a known best value."""

  def __init__(self, best_category: str = 'corner', *,
               output_relative_error: bool = False):
    if best_category not in _CATEGORIES:
      raise ValueError(f'best_category must be one of {_CATEGORIES}')
    self._best_category = best_category
    del output_relative_error
    problem = vz.ProblemStatement()
    root = problem.search_space.root
    root.add_categorical_param('categorical', list(_CATEGORIES))
    root.add_discrete_param('discrete', [-0.8, 0.0, 0.7])
    root.add_int_param('int', -2, 2)
    root.add_float_param('float', -1.0, 1.0)
    problem.metric_information.append(vz.MetricInformation(
        name='value', goal=vz.ObjectiveMetricGoal.MAXIMIZE))
    self._problem = problem

  @property
  def optimal_value(self) -> float:
    # category bonus 2 + discrete (-0.8)^2 + int (2/2)^2 + float 1.
    return 2.0 + 0.64 + 1.0 + 1.0

  def _value(self, trial: vz.Trial) -> float:
    cat = trial.parameters.get_value('categorical')
    disc = float(trial.parameters.get_value('discrete'))
    intv = float(trial.parameters.get_value('int'))
    flt = float(trial.parameters.get_value('float'))
    value = 2.0 if cat == self._best_category else 0.0
    value += disc * disc
    value += (intv / 2.0) ** 2
    value += flt * flt
    return value

  def evaluate(self, suggestions: Sequence[vz.Trial]) -> None:
    for trial in suggestions:
      trial.complete(vz.Measurement(metrics={'value': self._value(trial)}))

  def problem_statement(self) -> vz.ProblemStatement:
    return copy.deepcopy(self._problem)
