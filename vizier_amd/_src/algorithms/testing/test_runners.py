"""Designer smoke/convergence harnesses.

Capability parity with vizier/_src/algorithms/testing/test_runners.py
(RandomMetricsRunner :32,137): drive any designer through the
ask-evaluate-tell loop with random (or supplied) metrics and return the
completed trials, validating feasibility along the way.
"""

from __future__ import annotations

from typing import Callable, List, Optional, Sequence

import numpy as np

from vizier_amd import pyvizier as vz
from vizier_amd._src.algorithms.core.abstractions import (
    ActiveTrials,
    CompletedTrials,
    Designer,
)


class RandomMetricsRunner:
  """Runs a designer against random metric values."""

  def __init__(self, problem: vz.ProblemStatement, *, iters: int = 5,
               batch_size: int = 1, seed: Optional[int] = None,
               verify_parameters: bool = True,
               validate_metrics: bool = False):
    self._problem = problem
    self._iters = iters
    self._batch_size = batch_size
    self._rng = np.random.default_rng(seed)
    self._verify = verify_parameters
    self._validate_metrics = validate_metrics

  def run_designer(self, designer: Designer) -> List[vz.Trial]:
    trials: List[vz.Trial] = []
    uid = 0
    for _ in range(self._iters):
      suggestions = designer.suggest(self._batch_size)
      if not suggestions:
        break
      completed = []
      for s in suggestions:
        if self._verify:
          for pc in self._problem.search_space.parameters:
            value = s.parameters.get_value(pc.name, None)
            assert value is not None and pc.contains(value), (
                f'Infeasible suggestion for {pc.name}: {value!r}')
        uid += 1
        t = s.to_trial(uid)
        metrics = {mi.name: float(self._rng.standard_normal())
                   for mi in self._problem.metric_information}
        t.complete(vz.Measurement(metrics=metrics))
        completed.append(t)
      designer.update(CompletedTrials(completed), ActiveTrials())
      trials.extend(completed)
    return trials


def run_with_objective(designer: Designer, problem: vz.ProblemStatement,
                       objective: Callable[[vz.TrialSuggestion], float],
                       *, iters: int, batch_size: int = 1
                       ) -> List[vz.Trial]:
  """Drives a designer against a real objective; returns all trials."""
  metric_name = problem.metric_information.item().name
  trials: List[vz.Trial] = []
  uid = 0
  for _ in range(iters):
    for s in designer.suggest(batch_size):
      uid += 1
      t = s.to_trial(uid)
      t.complete(vz.Measurement(metrics={metric_name: objective(s)}))
      trials.append(t)
      designer.update(CompletedTrials([t]), ActiveTrials())
  return trials
