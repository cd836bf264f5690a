"""Meta-learning of designer hyperparameters.

Capability parity with
vizier/_src/algorithms/designers/meta_learning/meta_learning.py
(MetaLearningDesigner :98): an outer meta-designer tunes the inner
designer's hyperparameters; every `num_trials_per_update` completed
trials, the recent improvement is reported to the meta-designer as the
reward for the current hyperparameter setting and a new setting is
requested.
"""

from __future__ import annotations

import dataclasses
import math
from typing import Any, Callable, Dict, List, Optional, Sequence

from vizier_amd import pyvizier as vz
from vizier_amd._src.algorithms.core.abstractions import (
    ActiveTrials,
    CompletedTrials,
    Designer,
)

TunedDesignerFactory = Callable[[vz.ProblemStatement, Dict[str, Any]],
                                Designer]
MetaDesignerFactory = Callable[[vz.ProblemStatement], Designer]


@dataclasses.dataclass
class MetaLearningConfig:
  num_trials_per_update: int = 20
  meta_metric_name: str = 'meta_reward'


class MetaLearningDesigner(Designer):
  """Tunes inner-designer hyperparameters online."""

  def __init__(self, problem: vz.ProblemStatement,
               tuned_designer_factory: TunedDesignerFactory,
               meta_search_space: vz.SearchSpace,
               meta_designer_factory: Optional[MetaDesignerFactory] = None,
               config: Optional[MetaLearningConfig] = None):
    self._problem = problem
    self._config = config or MetaLearningConfig()
    self._tuned_factory = tuned_designer_factory
    meta_problem = vz.ProblemStatement(
        search_space=meta_search_space,
        metric_information=[vz.MetricInformation(
            name=self._config.meta_metric_name,
            goal=vz.ObjectiveMetricGoal.MAXIMIZE)])
    if meta_designer_factory is None:
      from vizier_amd._src.algorithms.designers.random import (
          RandomDesigner,
      )
      meta_designer_factory = lambda p: RandomDesigner(p.search_space,
                                                       seed=0)
    self._meta_problem = meta_problem
    self._meta_designer = meta_designer_factory(meta_problem)
    self._meta_trial_id = 0
    self._current_hparams = self._ask_meta()
    self._inner = tuned_designer_factory(problem,
                                         self._current_hparams)
    self._trials_since_update = 0
    self._best_before = -math.inf
    self._best = -math.inf
    self._all_completed: List[vz.Trial] = []

  def _ask_meta(self) -> Dict[str, Any]:
    suggestion = self._meta_designer.suggest(1)[0]
    self._meta_suggestion = suggestion
    return dict(suggestion.parameters.as_dict())

  def _tell_meta(self, reward: float) -> None:
    self._meta_trial_id += 1
    trial = self._meta_suggestion.to_trial(self._meta_trial_id)
    trial.complete(vz.Measurement(
        metrics={self._config.meta_metric_name: reward}))
    self._meta_designer.update(CompletedTrials([trial]), ActiveTrials())

  @property
  def current_hyperparameters(self) -> Dict[str, Any]:
    return dict(self._current_hparams)

  def update(self, completed: CompletedTrials, all_active: ActiveTrials
             ) -> None:
    for t in completed.trials:
      if t.final_measurement and not t.infeasible:
        value = next(iter(t.final_measurement.metrics.values())).value
        self._best = max(self._best, value)
    self._all_completed.extend(completed.trials)
    self._trials_since_update += len(completed.trials)
    self._inner.update(completed, all_active)

    if self._trials_since_update >= self._config.num_trials_per_update:
      # Reward = did this hyperparameter epoch improve the incumbent?
      reward = 1.0 if self._best > self._best_before else 0.0
      self._tell_meta(reward)
      self._best_before = self._best
      self._trials_since_update = 0
      self._current_hparams = self._ask_meta()
      self._inner = self._tuned_factory(self._problem,
                                        self._current_hparams)
      self._inner.update(CompletedTrials(self._all_completed),
                         all_active)

  def suggest(self, count: Optional[int] = None
              ) -> Sequence[vz.TrialSuggestion]:
    return self._inner.suggest(count)
