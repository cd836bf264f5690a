"""Per-type Firefly/Eagle-Strategy utilities.

Capability parity with
vizier/_src/algorithms/designers/eagle_strategy/eagle_strategy_utils.py
(FireflyAlgorithmConfig :35, Firefly :81, EagleStrategyUtils :103,
FireflyPool :437): per-parameter-TYPE visibility and pull forces,
categorical Bernoulli mixing, Laplace perturbation direction with
per-type perturbation scales, pool-capacity heuristic, exploration-rate
accentuation, and the penalize / stuck-detection / removal rules.

Internal representation: every numeric parameter value is kept in its
SCALED [0,1] space (the converter's per-parameter scaling), categorical
values as raw category strings — mixing and perturbation operate in the
scaled space and are unscaled + rounded back to feasible values only
when a suggestion materializes.
"""

from __future__ import annotations

import copy
import dataclasses
import math
from typing import Dict, List, Optional, Union

import numpy as np

from vizier_amd import pyvizier as vz
from vizier_amd.converters import core as converter_core

ParamValue = Union[float, str]


@dataclasses.dataclass
class FireflyAlgorithmConfig:
  """Hyperparameters (reference eagle_strategy_utils.py:35-79)."""

  gravity: float = 1.0
  negative_gravity: float = 0.02
  visibility: float = 3.0
  categorical_visibility: float = 0.2
  discrete_visibility: float = 1.0
  perturbation: float = 1e-1
  perturbation_lower_bound: float = 1e-3
  categorical_perturbation_factor: float = 25.0
  discrete_perturbation_factor: float = 10.0
  pure_categorical_perturbation: float = 0.1
  max_perturbation: float = 0.5
  penalize_factor: float = 0.9
  explore_rate: float = 1.0
  infeasible_force_factor: float = 0.0
  pool_size_factor: float = 1.2
  max_pool_size: int = 1000


@dataclasses.dataclass
class Firefly:
  """One pool member: its best trial (scaled values) + exploration state."""

  id_: int
  perturbation: float
  generation: int
  values: Dict[str, ParamValue]     # scaled numeric / raw categorical
  reward: float                     # maximize; -inf until measured
  infeasible: bool = False

  def to_json(self) -> dict:
    return {'id': self.id_, 'perturbation': self.perturbation,
            'generation': self.generation, 'values': self.values,
            'reward': None if not math.isfinite(self.reward)
            else self.reward,
            'infeasible': self.infeasible}

  @classmethod
  def from_json(cls, d: dict) -> 'Firefly':
    return cls(d['id'], d['perturbation'], d['generation'],
               dict(d['values']),
               -math.inf if d['reward'] is None else d['reward'],
               d.get('infeasible', False))


class EagleUtils:
  """Per-type parameter operations over a (flattened) search space."""

  def __init__(self, problem: vz.ProblemStatement,
               config: FireflyAlgorithmConfig,
               rng: np.random.Generator):
    self.config = config
    self.rng = rng
    self._configs: List[vz.ParameterConfig] = []
    for top in problem.search_space.parameters:
      self._configs.extend(top.traverse())
    self.n_parameters = len(self._configs)
    # Degrees of freedom per type: parameters with > 1 feasible value
    # (eagle_strategy_utils.py:187).
    self._dof: Dict[vz.ParameterType, int] = {}
    for cfg in self._configs:
      if self._num_feasible(cfg) > 1:
        self._dof[cfg.type] = self._dof.get(cfg.type, 0) + 1
    goal = list(problem.metric_information)[0].goal
    self._maximize = goal.is_maximize

  @staticmethod
  def _num_feasible(cfg: vz.ParameterConfig) -> int:
    if cfg.type == vz.ParameterType.CATEGORICAL:
      return len(cfg.feasible_values)
    if cfg.type == vz.ParameterType.DISCRETE:
      return len(cfg.feasible_values)
    if cfg.type == vz.ParameterType.INTEGER:
      lo, hi = cfg.bounds
      return int(hi - lo) + 1
    lo, hi = cfg.bounds
    return 2 if hi > lo else 1

  @property
  def parameter_configs(self) -> List[vz.ParameterConfig]:
    return list(self._configs)

  def is_pure_categorical(self) -> bool:
    return all(c.type == vz.ParameterType.CATEGORICAL
               for c in self._configs)

  def pool_capacity(self) -> int:
    """min(10 + round((df^1.2 + df) * 0.5), max_pool_size) (:235)."""
    df = self.n_parameters
    return min(10 + round((df ** self.config.pool_size_factor + df) * 0.5),
               self.config.max_pool_size)

  # -- scaled-value plumbing -------------------------------------------------

  def trial_to_values(self, trial: vz.TrialSuggestion
                      ) -> Dict[str, ParamValue]:
    out: Dict[str, ParamValue] = {}
    for cfg in self._configs:
      raw = trial.parameters.get_value(cfg.name, None)
      if cfg.type == vz.ParameterType.CATEGORICAL:
        out[cfg.name] = raw if raw is not None else cfg.feasible_values[0]
      else:
        out[cfg.name] = (converter_core._scale(cfg, float(raw))
                         if raw is not None else 0.5)
    return out

  def values_to_parameters(self, values: Dict[str, ParamValue]
                           ) -> vz.ParameterDict:
    params = vz.ParameterDict()
    for cfg in self._configs:
      v = values[cfg.name]
      if cfg.type == vz.ParameterType.CATEGORICAL:
        params[cfg.name] = v
      else:
        unscaled = converter_core._unscale(cfg, float(v))
        if cfg.type in (vz.ParameterType.INTEGER,
                        vz.ParameterType.DISCRETE):
          unscaled = cfg.round_to_feasible(unscaled)
        params[cfg.name] = unscaled
    return params

  def trial_reward(self, trial: vz.Trial, metric_name: str) -> float:
    if trial.infeasible or trial.final_measurement is None:
      return -math.inf
    metric = trial.final_measurement.metrics.get(metric_name)
    if metric is None:
      return -math.inf
    return float(metric.value) if self._maximize else -float(metric.value)

  # -- per-type forces (eagle_strategy_utils.py:132-233) ---------------------

  def distance_squared_by_type(self, v1: Dict[str, ParamValue],
                               v2: Dict[str, ParamValue]
                               ) -> Dict[vz.ParameterType, float]:
    out = {t: 0.0 for t in (vz.ParameterType.DOUBLE,
                            vz.ParameterType.DISCRETE,
                            vz.ParameterType.INTEGER,
                            vz.ParameterType.CATEGORICAL)}
    for cfg in self._configs:
      a, b = v1[cfg.name], v2[cfg.name]
      if cfg.type == vz.ParameterType.CATEGORICAL:
        # Reference counts EQUAL categories (:216) — matching values
        # increase the distance-squared term, which weakens the pull.
        out[cfg.type] += float(a == b)
      else:
        d = float(a) - float(b)   # already range-normalized (scaled)
        out[cfg.type] += d * d
    return out

  def canonical_distance(self, v1: Dict[str, ParamValue],
                         v2: Dict[str, ParamValue]) -> float:
    return sum(self.distance_squared_by_type(v1, v2).values())

  def pull_weights_by_type(self, other: Dict[str, ParamValue],
                           current: Dict[str, ParamValue],
                           other_is_better: bool
                           ) -> Dict[vz.ParameterType, float]:
    cfg = self.config
    d2 = self.distance_squared_by_type(other, current)
    direction = cfg.gravity if other_is_better else -cfg.negative_gravity
    vis = {vz.ParameterType.DOUBLE: cfg.visibility,
           vz.ParameterType.CATEGORICAL: cfg.categorical_visibility,
           vz.ParameterType.DISCRETE: cfg.discrete_visibility,
           vz.ParameterType.INTEGER: cfg.discrete_visibility}
    out = {}
    for ptype, dist2 in d2.items():
      dof = self._dof.get(ptype, 0)
      if dof == 0:
        out[ptype] = 0.0
        continue
      scaled = dist2 / dof * 10.0
      out[ptype] = math.exp(-vis[ptype] * scaled) * direction
    return out

  def explore_weight(self, w: float) -> float:
    """Exploration-rate accentuation (eagle_strategy.py:305-311)."""
    rate = self.config.explore_rate
    if w > 0.5:
      return rate * w + (1.0 - rate) * 1.0
    return rate * w

  def combine(self, cfg: vz.ParameterConfig, other: ParamValue,
              current: ParamValue, other_weight: float) -> ParamValue:
    """Weighted mix (eagle_strategy_utils.py:243-295), scaled space."""
    if cfg.type == vz.ParameterType.CATEGORICAL:
      if 0.0 < other_weight < 1.0:
        return other if self.rng.random() < other_weight else current
      return other if other_weight >= 1.0 else current
    mixed = float(other) * other_weight + float(current) * \
        (1.0 - other_weight)
    return min(max(mixed, 0.0), 1.0)

  # -- perturbation (eagle_strategy_utils.py:297-356) ------------------------

  def perturbation_scales(self) -> np.ndarray:
    scales = np.ones(self.n_parameters)
    for i, cfg in enumerate(self._configs):
      if cfg.type == vz.ParameterType.CATEGORICAL:
        scales[i] = self.config.categorical_perturbation_factor
      elif cfg.type == vz.ParameterType.DISCRETE:
        scales[i] = self.config.discrete_perturbation_factor / (
            max(self._num_feasible(cfg), 1) * self.config.perturbation)
    return scales

  def create_perturbations(self, perturbation: float) -> np.ndarray:
    if self.is_pure_categorical():
      return np.full(self.n_parameters,
                     self.config.pure_categorical_perturbation)
    noise = self.rng.laplace(size=self.n_parameters)
    direction = noise / max(np.max(np.abs(noise)), 1e-12)
    return direction * perturbation * self.perturbation_scales()

  def perturb(self, cfg: vz.ParameterConfig, value: ParamValue,
              amount: float) -> ParamValue:
    if cfg.type == vz.ParameterType.CATEGORICAL:
      if self.rng.random() < abs(amount):
        return cfg.feasible_values[
            self.rng.integers(len(cfg.feasible_values))]
      return value
    return min(max(float(value) + amount, 0.0), 1.0)


class FireflyPool:
  """Pool bookkeeping (eagle_strategy_utils.py:437-596)."""

  def __init__(self, utils: EagleUtils, capacity: int):
    self._utils = utils
    self.capacity = capacity
    self._pool: Dict[int, Firefly] = {}
    self._last_id = -1
    self._max_fly_id = 0

  @property
  def size(self) -> int:
    return len(self._pool)

  @property
  def members(self) -> Dict[int, Firefly]:
    return self._pool

  def generate_new_fly_id(self) -> int:
    self._max_fly_id += 1
    return self._max_fly_id - 1

  def remove_fly(self, fly: Firefly) -> None:
    self._pool.pop(fly.id_, None)

  def find_parent_fly(self, fly_id: Optional[int]) -> Optional[Firefly]:
    return self._pool.get(fly_id) if fly_id is not None else None

  def get_shuffled_flies(self, rng: np.random.Generator) -> List[Firefly]:
    flies = [copy.deepcopy(f) for f in self._pool.values()]
    rng.shuffle(flies)
    return flies

  def get_next_moving_fly_copy(self) -> Firefly:
    """Round-robin by id, skipping removed/infeasible flies (:482)."""
    curr = self._last_id + 1
    while curr != self._last_id:
      if curr > self._max_fly_id:
        curr = next(iter(self._pool))
      if curr in self._pool and not self._pool[curr].infeasible:
        self._last_id = curr
        return copy.deepcopy(self._pool[curr])
      curr += 1
    return copy.deepcopy(self._pool[self._last_id])

  def is_best_fly(self, fly: Firefly) -> bool:
    return all(other.id_ == fly.id_ or not (other.reward > fly.reward)
               for other in self._pool.values())

  def find_closest_parent(self, values: Dict[str, ParamValue]
                          ) -> Optional[Firefly]:
    best, best_d = None, float('inf')
    for fly in self._pool.values():
      if fly.infeasible:
        continue
      d = self._utils.canonical_distance(fly.values, values)
      if d < best_d:
        best, best_d = fly, d
    return best

  def create_or_update_fly(self, fly_id: int,
                           values: Dict[str, ParamValue], reward: float,
                           infeasible: bool) -> None:
    if fly_id not in self._pool:
      self._pool[fly_id] = Firefly(
          id_=fly_id, perturbation=self._utils.config.perturbation,
          generation=1, values=values, reward=reward,
          infeasible=infeasible)
    elif reward > self._pool[fly_id].reward:
      self._pool[fly_id].values = values
      self._pool[fly_id].reward = reward

  # -- serialization ---------------------------------------------------------

  def to_json(self) -> dict:
    return {'last_id': self._last_id, 'max_fly_id': self._max_fly_id,
            'capacity': self.capacity,
            'pool': [f.to_json() for f in self._pool.values()]}

  def load_json(self, d: dict) -> None:
    self._last_id = d['last_id']
    self._max_fly_id = d['max_fly_id']
    self.capacity = d['capacity']
    self._pool = {f['id']: Firefly.from_json(f) for f in d['pool']}
