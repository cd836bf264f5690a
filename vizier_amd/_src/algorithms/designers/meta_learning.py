"""Meta-learning of designer hyperparameters.

Capability parity with
vizier/_src/algorithms/designers/meta_learning/meta_learning.py
(MetaLearningDesigner, MetaLearningConfig, MetaLearningState) and
meta_learning_utils.py: a three-state process — INITIALIZE (run the
tuned designer with the search space's DEFAULT hyperparameter values
until `tuning_min_num_trials` trials accumulate), TUNE (every
`num_trials_per_tuning` trials, complete a meta-trial whose score is
the best tuned-trial objective of the epoch, ask the meta-designer for
new hyperparameters, rebuild the tuned designer and replay ALL trials
into it), and USE_BEST_PARAMS (past `tuning_max_num_trials`, lock in
the best meta-trial's hyperparameters). The meta problem inherits the
tuned problem's goal; infeasible trials never win.

`meta_eagle_search_space()` reproduces the reference's Eagle
hyperparameter space (eagle_meta_learning.py:23-108).
"""

from __future__ import annotations

import dataclasses
import enum
from typing import Any, Callable, Dict, List, Optional, Sequence

import numpy as np

from vizier_amd import pyvizier as vz
from vizier_amd._src.algorithms.core.abstractions import (
    ActiveTrials,
    CompletedTrials,
    Designer,
)

# Called as factory(problem, seed=..., **hyperparams) -> Designer
# (reference meta_learning.py:166-171).
TunedDesignerFactory = Callable[..., Designer]
MetaDesignerFactory = Callable[..., Designer]

_META_METRIC = 'score'


@dataclasses.dataclass
class MetaLearningConfig:
  """Reference defaults (meta_learning.py:58-79)."""

  num_trials_per_tuning: int = 100
  tuning_min_num_trials: int = 3000
  tuning_max_num_trials: int = 10000


class MetaLearningState(enum.Enum):
  INITIALIZE = 1
  TUNE = 2
  USE_BEST_PARAMS = 3


class MetaLearningDesigner(Designer):
  """Tunes inner-designer hyperparameters through a meta-designer."""

  def __init__(self, problem: vz.ProblemStatement,
               tuned_designer_factory: TunedDesignerFactory,
               tuning_hyperparams: vz.SearchSpace,
               meta_designer_factory: Optional[MetaDesignerFactory] = None,
               config: Optional[MetaLearningConfig] = None,
               seed: Optional[int] = None):
    metrics = list(problem.metric_information)
    if len(metrics) != 1:
      raise ValueError(f'Expected exactly one metric, got {len(metrics)}.')
    self._problem = problem
    self._goal = metrics[0].goal
    self._tuned_metric_name = metrics[0].name
    self._config = config or MetaLearningConfig()
    self._tuned_factory = tuned_designer_factory
    self._seed = (seed if seed is not None
                  else int(np.random.randint(0, 10 ** 6)))

    self._meta_problem = vz.ProblemStatement(
        search_space=tuning_hyperparams,
        metric_information=[vz.MetricInformation(
            name=_META_METRIC, goal=self._goal)])
    if meta_designer_factory is None:
      from vizier_amd._src.algorithms.designers.quasi_random import (
          QuasiRandomDesigner,
      )
      meta_designer_factory = (
          lambda p, seed=None: QuasiRandomDesigner(p.search_space,
                                                   seed=seed))
    self._meta_designer = meta_designer_factory(self._meta_problem,
                                                seed=self._seed)

    self._state = MetaLearningState.INITIALIZE
    self._meta_trials: List[vz.Trial] = []
    self._meta_trial_id = 0
    self._trials: List[vz.Trial] = []
    self._curr_trials: List[vz.Trial] = []
    self._curr_hyperparams = self._default_hyperparameters()
    self._curr_designer = self._build_tuned(self._curr_hyperparams)

  # -- helpers ---------------------------------------------------------------

  def _default_hyperparameters(self) -> vz.TrialSuggestion:
    """Defaults from the tuning space (meta_learning_utils.py:57-67)."""
    suggestion = vz.TrialSuggestion()
    for top in self._meta_problem.search_space.parameters:
      for cfg in top.traverse():
        if cfg.default_value is None:
          raise ValueError(
              f'Hyper-param {cfg.name!r} has no default value.')
        suggestion.parameters[cfg.name] = cfg.default_value
    return suggestion

  def _build_tuned(self, hyperparams: vz.TrialSuggestion) -> Designer:
    return self._tuned_factory(self._problem, seed=self._seed,
                               **dict(hyperparams.parameters.as_dict()))

  def _trial_score(self, trial: vz.Trial, metric_name: str
                   ) -> Optional[float]:
    if trial.infeasible or trial.final_measurement is None:
      return None
    metric = trial.final_measurement.metrics.get(metric_name)
    return None if metric is None else float(metric.value)

  def _best_trial(self, trials: Sequence[vz.Trial], metric_name: str
                  ) -> vz.Trial:
    def keyed(t):
      v = self._trial_score(t, metric_name)
      if v is None:
        return -float('inf')
      return v if self._goal.is_maximize else -v
    return max(trials, key=keyed)

  @property
  def state(self) -> MetaLearningState:
    return self._state

  @property
  def current_hyperparameters(self) -> Dict[str, Any]:
    return dict(self._curr_hyperparams.parameters.as_dict())

  # -- designer protocol ----------------------------------------------------

  def suggest(self, count: Optional[int] = None
              ) -> Sequence[vz.TrialSuggestion]:
    return self._curr_designer.suggest(count or 1)

  def update(self, completed: CompletedTrials, all_active: ActiveTrials
             ) -> None:
    """Reference state machine (meta_learning.py:185-259)."""
    cfg = self._config
    self._trials.extend(completed.trials)
    self._curr_trials.extend(completed.trials)
    self._curr_designer.update(CompletedTrials(completed.trials),
                               ActiveTrials())

    if len(self._trials) < cfg.tuning_min_num_trials:
      return

    if len(self._trials) >= cfg.tuning_max_num_trials:
      if self._state == MetaLearningState.TUNE and self._meta_trials:
        best = self._best_trial(self._meta_trials, _META_METRIC)
        self._curr_hyperparams = vz.TrialSuggestion(best.parameters)
        self._curr_designer = self._build_tuned(self._curr_hyperparams)
        self._curr_designer.update(CompletedTrials(self._trials),
                                   ActiveTrials())
        self._state = MetaLearningState.USE_BEST_PARAMS
      return

    self._state = MetaLearningState.TUNE
    if len(self._curr_trials) >= cfg.num_trials_per_tuning:
      # Complete the meta-trial: score = best tuned-trial value of the
      # epoch in the problem's own goal direction (continuous reward,
      # meta_learning_utils.py:82-85).
      best_epoch = self._best_trial(self._curr_trials,
                                    self._tuned_metric_name)
      score = self._trial_score(best_epoch, self._tuned_metric_name)
      self._meta_trial_id += 1
      meta_trial = self._curr_hyperparams.to_trial(self._meta_trial_id)
      if score is None:
        meta_trial.complete(vz.Measurement(), infeasibility_reason='all '
                            'epoch trials infeasible')
      else:
        meta_trial.complete(vz.Measurement(
            metrics={_META_METRIC: score}))
      self._meta_designer.update(CompletedTrials([meta_trial]),
                                 ActiveTrials())
      self._meta_trials.append(meta_trial)
      # New hyperparameters; replay every trial into a fresh designer.
      self._curr_hyperparams = self._meta_designer.suggest(1)[0]
      self._curr_designer = self._build_tuned(self._curr_hyperparams)
      self._curr_designer.update(CompletedTrials(self._trials),
                                 ActiveTrials())
      self._curr_trials = []


def meta_eagle_search_space() -> vz.SearchSpace:
  """Eagle hyperparameter tuning space (eagle_meta_learning.py:23-108)."""
  space = vz.SearchSpace()
  root = space.root
  log = vz.ScaleType.LOG
  root.add_float_param('perturbation', 1e-4, 1e2,
                       default_value=1e-1, scale_type=log)
  root.add_float_param('perturbation_lower_bound', 1e-5, 1e-1,
                       default_value=1e-3, scale_type=log)
  root.add_float_param('gravity', 1e-2, 1e2,
                       default_value=1.0, scale_type=log)
  root.add_float_param('visibility', 3e-2, 3e2,
                       default_value=3.0, scale_type=log)
  root.add_float_param('categorical_visibility', 2e-3, 2e1,
                       default_value=2e-1, scale_type=log)
  root.add_float_param('discrete_visibility', 1e-2, 1e2,
                       default_value=1.0, scale_type=log)
  root.add_float_param('categorical_perturbation_factor', 2.5e-1, 2.5e3,
                       default_value=2.5e1, scale_type=log)
  root.add_float_param('discrete_perturbation_factor', 1e-1, 1e3,
                       default_value=1e1, scale_type=log)
  root.add_float_param('pool_size_factor', 1.0, 2.0,
                       default_value=1.2, scale_type=log)
  root.add_float_param('negative_gravity', 2e-4, 2.0,
                       default_value=2e-2, scale_type=log)
  root.add_float_param('pure_categorical_perturbation', 1e-3, 1e1,
                       default_value=1e-1, scale_type=log)
  return space


def meta_eagle_designer_factory(problem: vz.ProblemStatement,
                                seed: Optional[int] = None,
                                **hyperparams) -> Designer:
  """Builds an EagleStrategyDesigner from meta-suggested hyperparams."""
  from vizier_amd._src.algorithms.designers.eagle_strategy.eagle_utils \
      import FireflyAlgorithmConfig
  from vizier_amd._src.algorithms.designers.eagle_strategy.eagle_strategy \
      import EagleStrategyDesigner
  return EagleStrategyDesigner(
      problem, FireflyAlgorithmConfig(**hyperparams), seed=seed)
