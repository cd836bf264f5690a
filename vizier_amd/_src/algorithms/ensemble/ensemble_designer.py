"""Ensembling designer: bandit over a set of expert designers.

Capability parity with
vizier/_src/algorithms/ensemble/ensemble_designer.py:110: each suggest
samples an expert via the EnsembleDesign strategy; rewards (objective
improvements) are attributed to the expert that produced the trial via
trial metadata.
"""

from __future__ import annotations

import math
from typing import Callable, Dict, Optional, Sequence

from vizier_amd import pyvizier as vz
from vizier_amd._src.algorithms.core.abstractions import (
    ActiveTrials,
    CompletedTrials,
    Designer,
)
from vizier_amd._src.algorithms.ensemble.ensemble_design import (
    EnsembleDesign,
    EXP3IXEnsembleDesign,
)

_NS = 'ensemble'
_EXPERT_KEY = 'expert'


class EnsembleDesigner(Designer):
  """Runs several designers, routing suggestions via a bandit."""

  def __init__(self, designers: Dict[str, Designer], *,
               strategy_factory: Callable[[Sequence[int]], EnsembleDesign]
               = EXP3IXEnsembleDesign,
               use_diversified_ranks: bool = False,
               seed: Optional[int] = None):
    if not designers:
      raise ValueError('Need at least one designer.')
    del use_diversified_ranks
    self._names = list(designers)
    self._designers = designers
    self._strategy = strategy_factory(list(range(len(self._names))))
    self._best_so_far = -math.inf

  def update(self, completed: CompletedTrials, all_active: ActiveTrials
             ) -> None:
    for trial in completed.trials:
      expert_name = trial.metadata.abs_ns((_NS,)).get(_EXPERT_KEY, None)
      if expert_name in self._names and trial.final_measurement and \
          not trial.infeasible:
        value = next(iter(trial.final_measurement.metrics.values())).value
        # Reward = normalized improvement over the incumbent.
        improved = 1.0 if value > self._best_so_far else 0.0
        self._best_so_far = max(self._best_so_far, value)
        self._strategy.update(self._names.index(expert_name), improved)
    for designer in self._designers.values():
      designer.update(completed, all_active)

  def suggest(self, count: Optional[int] = None
              ) -> Sequence[vz.TrialSuggestion]:
    count = count or 1
    out = []
    for _ in range(count):
      idx = self._strategy.sample()
      name = self._names[idx]
      suggestions = self._designers[name].suggest(1)
      for s in suggestions:
        s.metadata.abs_ns((_NS,))[_EXPERT_KEY] = name
        out.append(s)
    return out
