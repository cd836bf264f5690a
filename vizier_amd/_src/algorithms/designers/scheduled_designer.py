"""Time-scheduled designer hyperparameters.

Capability parity with
vizier/_src/algorithms/designers/scheduled_designer.py (ScheduledDesigner
:119, LinearScheduledParam :76) plus the scheduled GP-Bandit /
GP-UCB-PE instantiations (scheduled_gp_bandit.py,
scheduled_gp_ucb_pe.py): designer hyperparameters decay as a function of
expected-total-trials progress.
"""

from __future__ import annotations

import dataclasses
import math
from typing import Any, Callable, Dict, Optional, Sequence

from vizier_amd import pyvizier as vz
from vizier_amd._src.algorithms.core.abstractions import (
    ActiveTrials,
    CompletedTrials,
    Designer,
)


@dataclasses.dataclass
class LinearScheduledParam:
  init_value: float
  final_value: float

  def value(self, progress: float) -> float:
    progress = min(max(progress, 0.0), 1.0)
    return self.init_value + (self.final_value - self.init_value) * progress


@dataclasses.dataclass
class ExponentialScheduledParam:
  init_value: float
  final_value: float
  rate: float = 1.0

  def value(self, progress: float) -> float:
    progress = min(max(progress, 0.0), 1.0)
    if self.init_value <= 0 or self.final_value <= 0:
      raise ValueError('Exponential schedule requires positive endpoints.')
    log_v = (math.log(self.init_value) +
             (math.log(self.final_value) - math.log(self.init_value)) *
             progress ** self.rate)
    return math.exp(log_v)


class ScheduledDesigner(Designer):
  """Rebuilds the inner designer whenever scheduled params change.

  `designer_factory(problem, **params)` receives the scheduled values;
  progress = num_completed / expected_total_num_trials.
  """

  def __init__(self, problem: vz.ProblemStatement,
               designer_factory: Callable[..., Designer],
               scheduled_params: Dict[str, Any], *,
               expected_total_num_trials: int):
    self._problem = problem
    self._factory = designer_factory
    self._params = scheduled_params
    self._total = expected_total_num_trials
    self._num_completed = 0
    self._all_completed: list = []
    self._designer: Optional[Designer] = None

  def _progress(self) -> float:
    return self._num_completed / max(self._total, 1)

  def current_param_values(self) -> Dict[str, float]:
    p = self._progress()
    return {name: sched.value(p) for name, sched in self._params.items()}

  def update(self, completed: CompletedTrials, all_active: ActiveTrials
             ) -> None:
    self._num_completed += len(completed.trials)
    self._all_completed.extend(completed.trials)
    self._all_active = all_active

  def suggest(self, count: Optional[int] = None
              ) -> Sequence[vz.TrialSuggestion]:
    designer = self._factory(self._problem, **self.current_param_values())
    designer.update(CompletedTrials(self._all_completed),
                    getattr(self, '_all_active', ActiveTrials()))
    return designer.suggest(count)


def scheduled_gp_bandit(problem: vz.ProblemStatement, *,
                        expected_total_num_trials: int,
                        init_ucb_coefficient: float = 4.0,
                        final_ucb_coefficient: float = 1.0,
                        decay_rate: float = 1.2,
                        **gp_kwargs) -> ScheduledDesigner:
  """GP-Bandit with an exponentially decaying UCB coefficient
  (parity with scheduled_gp_bandit.py)."""
  from vizier_amd._src.algorithms.designers.gp_bandit import (
      GPBanditConfig,
      VizierGPBandit,
  )

  def factory(p, ucb_coefficient):
    return VizierGPBandit(p, GPBanditConfig(
        ucb_coefficient=ucb_coefficient, **gp_kwargs))

  return ScheduledDesigner(
      problem, factory,
      {'ucb_coefficient': ExponentialScheduledParam(
          init_ucb_coefficient, final_ucb_coefficient, decay_rate)},
      expected_total_num_trials=expected_total_num_trials)


def scheduled_gp_ucb_pe(problem: vz.ProblemStatement, *,
                        expected_total_num_trials: int,
                        init_ucb_coefficient: float = 4.0,
                        final_ucb_coefficient: float = 1.0,
                        decay_ucb_coefficient: float = 1.2,
                        init_explore_region_ucb_coefficient: float = 1.0,
                        final_explore_region_ucb_coefficient: float = 0.5,
                        decay_explore_region_ucb_coefficient: float = 1.2,
                        **pe_kwargs) -> ScheduledDesigner:
  """GP-UCB-PE with exponentially decaying UCB and explore-region
  coefficients (parity with scheduled_gp_ucb_pe.py:29
  ScheduledGPUCBPEFactory)."""
  from vizier_amd._src.algorithms.designers.gp_ucb_pe import (
      UCBPEConfig,
      VizierGPUCBPEBandit,
  )

  def factory(p, ucb_coefficient, explore_region_ucb_coefficient):
    cfg = UCBPEConfig(
        ucb_coefficient=ucb_coefficient,
        explore_region_ucb_coefficient=explore_region_ucb_coefficient,
        **pe_kwargs)
    return VizierGPUCBPEBandit(p, cfg)

  return ScheduledDesigner(
      problem, factory,
      {'ucb_coefficient': ExponentialScheduledParam(
          init_ucb_coefficient, final_ucb_coefficient,
          decay_ucb_coefficient),
       'explore_region_ucb_coefficient': ExponentialScheduledParam(
           init_explore_region_ucb_coefficient,
           final_explore_region_ucb_coefficient,
           decay_explore_region_ucb_coefficient)},
      expected_total_num_trials=expected_total_num_trials)
