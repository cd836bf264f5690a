"""NSGA-II multi-objective evolutionary designer.

Capability parity with vizier/_src/algorithms/evolution/nsga2.py
(NSGA2Survival :149, NSGA2Designer :244) and numpy_populations.py
(Population :94): non-dominated sorting + crowding-distance survival,
binary tournament selection, uniform crossover and L-inf-bounded
mutation over the converter's [0,1] feature space.
"""

from __future__ import annotations

import dataclasses
from typing import List, Optional, Sequence, Tuple

import numpy as np

from vizier_amd import pyvizier as vz
from vizier_amd.converters.core import TrialToArrayConverter
from vizier_amd._src.algorithms.core.abstractions import (
    ActiveTrials,
    CompletedTrials,
    Designer,
)
from vizier_amd._src.pyvizier import multimetric


def pareto_rank(ys: np.ndarray) -> np.ndarray:
  """0 = first front, 1 = dominated only by front 0, ... (maximize)."""
  n = len(ys)
  rank = np.full(n, -1, dtype=int)
  remaining = np.arange(n)
  front = 0
  while len(remaining):
    optimal = multimetric.is_pareto_optimal(ys[remaining])
    rank[remaining[optimal]] = front
    remaining = remaining[~optimal]
    front += 1
  return rank


def crowding_distance(ys: np.ndarray) -> np.ndarray:
  """Per-point crowding distance within its own set (maximize spread)."""
  n, m = ys.shape
  if n <= 2:
    return np.full(n, np.inf)
  dist = np.zeros(n)
  for j in range(m):
    order = np.argsort(ys[:, j])
    span = ys[order[-1], j] - ys[order[0], j]
    dist[order[0]] = dist[order[-1]] = np.inf
    if span <= 0:
      continue
    dist[order[1:-1]] += (ys[order[2:], j] - ys[order[:-2], j]) / span
  return dist


def nsga2_survival(ys: np.ndarray, pop_size: int) -> np.ndarray:
  """Indices of the surviving population (rank, then crowding)."""
  ranks = pareto_rank(ys)
  survivors: List[int] = []
  for front in range(ranks.max() + 1):
    members = np.flatnonzero(ranks == front)
    if len(survivors) + len(members) <= pop_size:
      survivors.extend(members.tolist())
    else:
      crowd = crowding_distance(ys[members])
      order = members[np.argsort(-crowd)]
      survivors.extend(order[:pop_size - len(survivors)].tolist())
      break
  return np.asarray(survivors, dtype=int)


@dataclasses.dataclass
class NSGA2Config:
  population_size: int = 50
  crossover_prob: float = 0.9
  mutation_prob_per_dim: Optional[float] = None  # default 1/D
  mutation_linf: float = 0.1  # max per-dim move in scaled space


class NSGA2Designer(Designer):
  """NSGA-II over the converter's scaled feature space."""

  def __init__(self, problem: vz.ProblemStatement,
               config: Optional[NSGA2Config] = None, *,
               seed: Optional[int] = None, population_size: Optional[int]
               = None, first_survival_after: Optional[int] = None):
    self._problem = problem
    self._config = config or NSGA2Config()
    if population_size is not None:
      self._config.population_size = population_size
    del first_survival_after  # API-parity arg; survival runs every update
    self._converter = TrialToArrayConverter(problem)
    self._rng = np.random.default_rng(seed)
    self._pop_x: Optional[np.ndarray] = None    # (P, D)
    self._pop_y: Optional[np.ndarray] = None    # (P, M)

  def update(self, completed: CompletedTrials, all_active: ActiveTrials
             ) -> None:
    del all_active
    trials = [t for t in completed.trials
              if t.final_measurement is not None and not t.infeasible]
    if not trials:
      return
    xs = self._converter.to_features(trials)
    ys = self._converter.to_labels(trials)
    keep = ~np.isnan(ys).any(axis=1)
    xs, ys = xs[keep], ys[keep]
    if not len(xs):
      return
    if self._pop_x is None:
      self._pop_x, self._pop_y = xs, ys
    else:
      self._pop_x = np.concatenate([self._pop_x, xs])
      self._pop_y = np.concatenate([self._pop_y, ys])
    if len(self._pop_x) > self._config.population_size:
      idx = nsga2_survival(self._pop_y.astype(np.float64),
                           self._config.population_size)
      self._pop_x = self._pop_x[idx]
      self._pop_y = self._pop_y[idx]

  def _tournament(self) -> np.ndarray:
    ranks = pareto_rank(self._pop_y.astype(np.float64))
    crowd = np.zeros(len(ranks))
    for front in range(ranks.max() + 1):
      members = np.flatnonzero(ranks == front)
      crowd[members] = crowding_distance(self._pop_y[members])
    a, b = self._rng.integers(0, len(self._pop_x), 2)
    if ranks[a] < ranks[b] or (ranks[a] == ranks[b] and
                               crowd[a] > crowd[b]):
      return self._pop_x[a]
    return self._pop_x[b]

  def suggest(self, count: Optional[int] = None
              ) -> Sequence[vz.TrialSuggestion]:
    count = count or 1
    d = self._converter.n_features
    cfg = self._config
    out = []
    for _ in range(count):
      if self._pop_x is None or len(self._pop_x) < 2:
        child = self._rng.uniform(0, 1, d)
      else:
        p1, p2 = self._tournament(), self._tournament()
        if self._rng.random() < cfg.crossover_prob:
          mask = self._rng.random(d) < 0.5
          child = np.where(mask, p1, p2).astype(np.float64)
        else:
          child = p1.astype(np.float64).copy()
        p_mut = cfg.mutation_prob_per_dim or (1.0 / max(d, 1))
        mutate = self._rng.random(d) < p_mut
        child[mutate] += self._rng.uniform(-cfg.mutation_linf,
                                           cfg.mutation_linf,
                                           mutate.sum())
        child = np.clip(child, 0.0, 1.0)
      out.append(vz.TrialSuggestion(
          self._converter.to_parameters(child[None, :])[0]))
    return out
