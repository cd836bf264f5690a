"""Safety wrapper: unsafe trials are shown to the inner designer as
infeasible.

Capability parity with
vizier/_src/algorithms/designers/unsafe_as_infeasible_designer.py:27.
"""

from __future__ import annotations

import copy
from typing import Callable, Optional, Sequence

from vizier_amd import pyvizier as vz
from vizier_amd._src.algorithms.core.abstractions import (
    ActiveTrials,
    CompletedTrials,
    Designer,
)
from vizier_amd._src.pyvizier.multimetric import SafetyChecker


class UnsafeAsInfeasibleDesigner(Designer):
  """Marks safety-violating trials infeasible before the inner update."""

  def __init__(self, problem: vz.ProblemStatement,
               designer_factory: Callable[[vz.ProblemStatement], Designer]):
    self._checker = SafetyChecker(problem.metric_information)
    self._designer = designer_factory(problem)

  def update(self, completed: CompletedTrials, all_active: ActiveTrials
             ) -> None:
    safe_flags = self._checker.are_trials_safe(completed.trials)
    converted = []
    for trial, safe in zip(completed.trials, safe_flags):
      if safe:
        converted.append(trial)
      else:
        clone = copy.deepcopy(trial)
        clone.complete(clone.final_measurement or vz.Measurement(),
                       infeasibility_reason='unsafe')
        converted.append(clone)
    self._designer.update(CompletedTrials(converted), all_active)

  def suggest(self, count: Optional[int] = None
              ) -> Sequence[vz.TrialSuggestion]:
    return self._designer.suggest(count)
