"""Adaptive bandit strategies for ensembling designers.

Capability parity with vizier/_src/algorithms/ensemble/ensemble_design.py
(:27-165): EXP3-Uniform, EXP3-IX, and an adaptive (Adapt-ML-Prod style)
strategy over expert (designer) indices, fed observed rewards.
"""

from __future__ import annotations

import abc
import dataclasses
from typing import List, Optional, Sequence

import numpy as np


@dataclasses.dataclass
class EnsembleDesignConfig:
  learning_rate: float = 1.0
  gamma: float = 0.1  # exploration mix / IX bias
  reward_range: tuple = (0.0, 1.0)


class EnsembleDesign(abc.ABC):
  """Maintains expert probabilities from (expert, reward) observations."""

  def __init__(self, indices: Sequence[int],
               config: Optional[EnsembleDesignConfig] = None, *,
               seed: Optional[int] = None):
    self.indices = list(indices)
    self.config = config or EnsembleDesignConfig()
    self._rng = np.random.default_rng(seed)
    self._rewards: List[tuple] = []  # (expert_index, reward)

  @property
  @abc.abstractmethod
  def ensemble_probs(self) -> np.ndarray:
    ...

  def sample(self) -> int:
    probs = self.ensemble_probs
    return int(self._rng.choice(self.indices, p=probs))

  def update(self, expert: int, reward: float) -> None:
    lo, hi = self.config.reward_range
    reward = (min(max(reward, lo), hi) - lo) / max(hi - lo, 1e-12)
    self._rewards.append((expert, reward))


class RandomEnsembleDesign(EnsembleDesign):

  @property
  def ensemble_probs(self) -> np.ndarray:
    k = len(self.indices)
    return np.full(k, 1.0 / k)


class EXP3UniformEnsembleDesign(EnsembleDesign):
  """EXP3 with uniform exploration mixing."""

  @property
  def ensemble_probs(self) -> np.ndarray:
    k = len(self.indices)
    gains = np.zeros(k)
    for t, (expert, reward) in enumerate(self._rewards):
      probs = self._probs_from_gains(gains)
      i = self.indices.index(expert)
      gains[i] += reward / max(probs[i], 1e-12)
    return self._probs_from_gains(gains)

  def _probs_from_gains(self, gains: np.ndarray) -> np.ndarray:
    k = len(gains)
    lr = self.config.learning_rate / max(k, 1)
    w = np.exp(lr * (gains - gains.max()))
    probs = w / w.sum()
    return (1 - self.config.gamma) * probs + self.config.gamma / k


class EXP3IXEnsembleDesign(EnsembleDesign):
  """EXP3-IX: implicit exploration via the gamma bias in the estimator."""

  @property
  def ensemble_probs(self) -> np.ndarray:
    k = len(self.indices)
    losses = np.zeros(k)
    for expert, reward in self._rewards:
      probs = self._probs_from_losses(losses)
      i = self.indices.index(expert)
      losses[i] += (1.0 - reward) / (probs[i] + self.config.gamma)
    return self._probs_from_losses(losses)

  def _probs_from_losses(self, losses: np.ndarray) -> np.ndarray:
    lr = self.config.learning_rate / max(len(losses), 1)
    w = np.exp(-lr * (losses - losses.min()))
    return w / w.sum()


class AdaptiveEnsembleDesign(EnsembleDesign):
  """Adapt-ML-Prod-style: per-expert adaptive learning rates."""

  def __init__(self, indices: Sequence[int],
               config: Optional[EnsembleDesignConfig] = None, *,
               max_lr: float = 0.5, seed: Optional[int] = None):
    super().__init__(indices, config, seed=seed)
    self._max_lr = max_lr

  @property
  def ensemble_probs(self) -> np.ndarray:
    k = len(self.indices)
    log_w = np.zeros(k)
    var = np.zeros(k)
    for expert, reward in self._rewards:
      w = np.exp(log_w - log_w.max())
      probs = w / w.sum()
      i = self.indices.index(expert)
      est = np.zeros(k)
      est[i] = reward / max(probs[i], 1e-12)
      instant = est - (probs @ est)
      lrs = np.minimum(self._max_lr, 1.0 / (1.0 + np.sqrt(var)))
      log_w = log_w + lrs * instant - lrs ** 2 * instant ** 2
      var = var + instant ** 2
    w = np.exp(log_w - log_w.max())
    probs = w / w.sum()
    return (1 - self.config.gamma) * probs + self.config.gamma / k
