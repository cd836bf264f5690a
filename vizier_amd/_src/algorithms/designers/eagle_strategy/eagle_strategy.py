"""Pure-Python serializable Eagle Strategy (Firefly) designer.

Capability parity with
vizier/_src/algorithms/designers/eagle_strategy/eagle_strategy.py
(EagleStrategyDesigner :95) at the reference's behavioral depth:
per-parameter-TYPE pull forces and perturbations (eagle_utils.py —
reference eagle_strategy_utils.py:103-435), exploration-rate
accentuation, categorical Bernoulli mixing, stuck-fly detection
(identical parameters => perturbation x10 capped), capacity-gated
removal that never evicts the best fly, closest-parent adoption of
foreign trials, and full pool + RNG serialization into study metadata
(PartiallySerializableDesigner). Serves the EAGLE_STRATEGY algorithm
string; the GPU-vectorized variant used inside GP-Bandit lives in
vizier_amd/_src/algorithms/optimizers/eagle.py.
"""

from __future__ import annotations

import json
import math
from typing import Optional, Sequence

import numpy as np

from vizier_amd import pyvizier as vz
from vizier_amd._src.algorithms.core.abstractions import (
    ActiveTrials,
    CompletedTrials,
    PartiallySerializableDesigner,
)
from vizier_amd._src.algorithms.designers.eagle_strategy.eagle_utils import (
    EagleUtils,
    FireflyAlgorithmConfig,
    FireflyPool,
)

_NS = 'eagle'
_PARENT_KEY = 'parent_fly_id'


class EagleStrategyDesigner(PartiallySerializableDesigner):
  """Firefly pool evolved one suggestion at a time."""

  def __init__(self, problem: vz.ProblemStatement,
               config: Optional[FireflyAlgorithmConfig] = None, *,
               seed: Optional[int] = None):
    if problem.search_space.is_conditional:
      raise ValueError('EagleStrategyDesigner does not support '
                       'conditional search spaces.')
    metrics = list(problem.metric_information)
    if len(metrics) != 1:
      raise ValueError('EagleStrategyDesigner is single-objective.')
    self._problem = problem
    self._metric_name = metrics[0].name
    self._config = config or FireflyAlgorithmConfig()
    self._rng = np.random.default_rng(seed)
    self._utils = EagleUtils(problem, self._config, self._rng)
    self._firefly_pool = FireflyPool(self._utils,
                                     self._utils.pool_capacity())
    from vizier_amd._src.algorithms.designers.quasi_random import (
        QuasiRandomDesigner,
    )
    self._initial_designer = QuasiRandomDesigner(problem.search_space,
                                                 seed=seed)

  # Back-compat attribute used by tests/PARITY checks.
  @property
  def _pool(self):
    return self._firefly_pool.members

  # -- designer protocol ----------------------------------------------------

  def suggest(self, count: Optional[int] = None
              ) -> Sequence[vz.TrialSuggestion]:
    return [self._suggest_one() for _ in range(count or 1)]

  def _suggest_one(self) -> vz.TrialSuggestion:
    if self._firefly_pool.size < self._firefly_pool.capacity:
      # Underpopulated pool: quasi-random init (reference seeds with a
      # serializable initial designer, eagle_strategy.py:155-158).
      params = self._initial_designer.suggest(1)[0].parameters
      parent_id = self._firefly_pool.generate_new_fly_id()
      suggestion = vz.TrialSuggestion(params)
    else:
      moving = self._firefly_pool.get_next_moving_fly_copy()
      self._mutate_fly(moving)
      self._perturb_fly(moving)
      parent_id = moving.id_
      suggestion = vz.TrialSuggestion(
          self._utils.values_to_parameters(moving.values))
    suggestion.metadata.ns(_NS)[_PARENT_KEY] = str(parent_id)
    return suggestion

  def _mutate_fly(self, moving) -> None:
    """Applies pulls from every pool member in shuffled order."""
    for other in self._firefly_pool.get_shuffled_flies(self._rng):
      other_better = other.reward > moving.reward
      weights = self._utils.pull_weights_by_type(
          other.values, moving.values, other_better)
      for cfg in self._utils.parameter_configs:
        w = weights[cfg.type]
        if other.infeasible:
          w *= self._config.infeasible_force_factor
        w = self._utils.explore_weight(w)
        moving.values[cfg.name] = self._utils.combine(
            cfg, other.values[cfg.name], moving.values[cfg.name], w)

  def _perturb_fly(self, moving) -> None:
    amounts = self._utils.create_perturbations(moving.perturbation)
    for i, cfg in enumerate(self._utils.parameter_configs):
      moving.values[cfg.name] = self._utils.perturb(
          cfg, moving.values[cfg.name], float(amounts[i]))

  def update(self, completed: CompletedTrials, all_active: ActiveTrials
             ) -> None:
    del all_active
    for trial in completed.trials:
      self._update_one(trial)

  def _update_one(self, trial: vz.Trial) -> None:
    values = self._utils.trial_to_values(trial)
    reward = self._utils.trial_reward(trial, self._metric_name)
    parent_raw = trial.metadata.abs_ns((_NS,)).get(_PARENT_KEY, None)
    if parent_raw is None:
      # Foreign trial: assign a fresh id (reference eagle_strategy.py
      # :365-369).
      parent_id = self._firefly_pool.generate_new_fly_id()
    else:
      parent_id = int(parent_raw)

    if trial.infeasible and self._config.infeasible_force_factor > 0:
      self._firefly_pool.create_or_update_fly(
          self._firefly_pool.generate_new_fly_id(), values, -math.inf,
          infeasible=True)

    parent = self._firefly_pool.find_parent_fly(parent_id)
    if parent is None:
      if trial.infeasible:
        return
      if self._firefly_pool.size < self._firefly_pool.capacity:
        self._firefly_pool.create_or_update_fly(parent_id, values,
                                                reward, infeasible=False)
        return
      # At capacity: adopt via the closest parent, but only when the
      # trial improves on it (otherwise the parent is not responsible
      # for the failure — eagle_strategy.py:407-437).
      closest = self._firefly_pool.find_closest_parent(values)
      if closest is None or not (reward > closest.reward):
        return
      parent = closest

    if not trial.infeasible and reward > parent.reward:
      parent.values = values
      parent.reward = reward
      parent.generation += 1
    else:
      self._penalize_parent(parent, values)

  def _penalize_parent(self, parent, values) -> None:
    """Reference eagle_strategy.py:438-464."""
    if values == parent.values:
      # Identical parameters: the fly is STUCK — escalate exploration.
      parent.perturbation = min(parent.perturbation * 10.0,
                                self._config.max_perturbation)
    else:
      parent.perturbation *= self._config.penalize_factor
    if parent.perturbation < self._config.perturbation_lower_bound:
      # Remove only at capacity and never the best fly (critical for
      # studies with few feasible trials).
      if (self._firefly_pool.size == self._firefly_pool.capacity and
          not self._firefly_pool.is_best_fly(parent)):
        self._firefly_pool.remove_fly(parent)

  # -- serialization --------------------------------------------------------

  def dump(self) -> vz.Metadata:
    md = vz.Metadata()
    state = {
        'rng': self._rng.bit_generator.state,
        'firefly_pool': self._firefly_pool.to_json(),
        'version': 'v2',
    }
    md.ns(_NS)['state'] = json.dumps(state)
    md.ns(_NS).ns('initial_designer').attach(
        self._initial_designer.dump())
    return md

  def load(self, metadata: vz.Metadata) -> None:
    blob = metadata.abs_ns((_NS,)).get('state', None)
    if blob is None:
      raise ValueError('No Eagle state found in metadata.')
    state = json.loads(blob)
    self._rng.bit_generator.state = state['rng']
    self._firefly_pool.load_json(state['firefly_pool'])
    try:
      self._initial_designer.load(
          metadata.ns(_NS).ns('initial_designer'))
    except Exception:
      pass  # harmless: the quasi-random stream restarts
