"""Pure-Python serializable Eagle Strategy (Firefly) designer.

Capability parity with
vizier/_src/algorithms/designers/eagle_strategy/eagle_strategy.py
(EagleStrategyDesigner :95, FireflyPool/EagleStrategyUtils
eagle_strategy_utils.py:103,437, serialization.py): a persistent firefly
pool evolved one suggestion at a time through the service protocol, with
full state (pool, perturbations, rewards, ids) serialized into study
metadata so the designer survives across Pythia calls. This serves the
EAGLE_STRATEGY algorithm string; the GPU-vectorized variant used inside
GP-Bandit lives in vizier_amd/_src/algorithms/optimizers/eagle.py.
"""

from __future__ import annotations

import json
import math
from typing import Dict, List, Optional, Sequence

import numpy as np

from vizier_amd import pyvizier as vz
from vizier_amd.converters.core import TrialToArrayConverter
from vizier_amd._src.algorithms.core.abstractions import (
    ActiveTrials,
    CompletedTrials,
    PartiallySerializableDesigner,
)
from vizier_amd._src.algorithms.optimizers.eagle import (
    EagleStrategyConfig,
    compute_pool_size,
)

_NS = 'eagle'
_FLY_ID_KEY = 'firefly_id'


class _Firefly:

  def __init__(self, fly_id: int, features: np.ndarray,
               reward: float = -math.inf, perturbation: float = 0.16):
    self.id = fly_id
    self.features = features            # dense converter features
    self.reward = reward
    self.perturbation = perturbation

  def to_json(self) -> dict:
    return {'id': self.id, 'features': self.features.tolist(),
            'reward': self.reward, 'perturbation': self.perturbation}

  @classmethod
  def from_json(cls, d: dict) -> '_Firefly':
    return cls(d['id'], np.asarray(d['features'], dtype=np.float64),
               d['reward'], d['perturbation'])


class EagleStrategyDesigner(PartiallySerializableDesigner):
  """Firefly pool updated incrementally from completed trials."""

  def __init__(self, problem: vz.ProblemStatement,
               config: Optional[EagleStrategyConfig] = None, *,
               seed: Optional[int] = None):
    self._problem = problem
    self._config = config or EagleStrategyConfig()
    self._converter = TrialToArrayConverter(problem)
    self._rng = np.random.default_rng(seed)
    d = self._converter.n_features
    self._pool_capacity = compute_pool_size(d, None, self._config)
    self._pool: Dict[int, _Firefly] = {}
    self._next_id = 0
    self._best_reward = -math.inf

  # -- designer protocol ----------------------------------------------------

  def update(self, completed: CompletedTrials, all_active: ActiveTrials
             ) -> None:
    del all_active
    for trial in completed.trials:
      self._update_one(trial)

  def _trial_reward(self, trial: vz.Trial) -> float:
    if trial.final_measurement is None or trial.infeasible:
      return -math.inf
    y = self._converter.to_labels([trial])[0, 0]
    return float(y) if np.isfinite(y) else -math.inf

  def _update_one(self, trial: vz.Trial) -> None:
    reward = self._trial_reward(trial)
    self._best_reward = max(self._best_reward, reward)
    features = self._converter.to_features([trial])[0].astype(np.float64)
    fly_id = trial.metadata.abs_ns((_NS,)).get(_FLY_ID_KEY, None)
    fly = self._pool.get(int(fly_id)) if fly_id is not None else None

    if fly is None:
      # Trial from elsewhere (or pre-pool): adopt it if there is room or
      # it beats the closest pool member.
      if len(self._pool) < self._pool_capacity:
        self._spawn(features, reward)
      else:
        closest = min(self._pool.values(), key=lambda f: float(
            np.sum((f.features - features) ** 2)))
        if reward > closest.reward:
          closest.features, closest.reward = features, reward
      return

    if reward > fly.reward:
      fly.features = features
      fly.reward = reward
    else:
      fly.perturbation *= self._config.penalize_factor
      if (fly.perturbation < self._config.perturbation_lower_bound and
          fly.reward != self._best_reward):
        # Dead fly: replace with a random restart.
        del self._pool[fly.id]
        self._spawn(self._random_features(), -math.inf)

  def _spawn(self, features: np.ndarray, reward: float) -> _Firefly:
    fly = _Firefly(self._next_id, features, reward,
                   self._config.perturbation)
    self._next_id += 1
    self._pool[fly.id] = fly
    return fly

  def _random_features(self) -> np.ndarray:
    d = self._converter.n_features
    return self._rng.uniform(0, 1, d)

  def suggest(self, count: Optional[int] = None
              ) -> Sequence[vz.TrialSuggestion]:
    count = count or 1
    out = []
    for _ in range(count):
      out.append(self._suggest_one())
    return out

  def _suggest_one(self) -> vz.TrialSuggestion:
    if len(self._pool) < self._pool_capacity:
      fly = self._spawn(self._random_features(), -math.inf)
      features = fly.features
    else:
      # Round-robin through the pool by id.
      ids = sorted(self._pool)
      fly = self._pool[ids[self._next_id % len(ids)]]
      self._next_id += 1
      features = self._mutate(fly)
    suggestion = vz.TrialSuggestion(
        self._converter.to_parameters(features[None, :])[0])
    suggestion.metadata.abs_ns((_NS,))[_FLY_ID_KEY] = str(fly.id)
    return suggestion

  def _mutate(self, fly: _Firefly) -> np.ndarray:
    cfg = self._config
    d = self._converter.n_features
    others = [f for f in self._pool.values()
              if f.id != fly.id and math.isfinite(f.reward)]
    x = fly.features.copy()
    pulls, pushes = [], []
    for other in others:
      d2 = float(np.sum((other.features - x) ** 2))
      force = math.exp(-cfg.visibility * d2 / max(d, 1) * 10.0)
      if other.reward >= fly.reward:
        pulls.append((cfg.gravity * force, other.features))
      else:
        pushes.append((-cfg.negative_gravity * force, other.features))
    move = np.zeros(d)
    for group in (pulls, pushes):
      if group:
        for s, feat in group:
          move += (cfg.normalization_scale * s / len(group)) * (feat - x)
    noise = self._rng.laplace(size=d)
    noise = np.sign(noise) * fly.perturbation
    return np.clip(x + move + noise, 0.0, 1.0)

  # -- serialization --------------------------------------------------------

  def dump(self) -> vz.Metadata:
    state = {
        'next_id': self._next_id,
        'best_reward': (self._best_reward
                        if math.isfinite(self._best_reward) else None),
        'pool': [f.to_json() for f in self._pool.values()],
    }
    md = vz.Metadata()
    md.ns(_NS)['state'] = json.dumps(state)
    return md

  def load(self, metadata: vz.Metadata) -> None:
    blob = metadata.abs_ns((_NS,)).get('state', None)
    if blob is None:
      raise ValueError('No Eagle state found in metadata.')
    state = json.loads(blob)
    self._next_id = state['next_id']
    self._best_reward = (state['best_reward']
                         if state['best_reward'] is not None else -math.inf)
    self._pool = {f['id']: _Firefly.from_json(f) for f in state['pool']}
