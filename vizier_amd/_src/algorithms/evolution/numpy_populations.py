"""Array-backed populations for the evolution template.

Capability parity with
vizier/_src/algorithms/evolution/numpy_populations.py (Offspring :94,
Population :167, PopulationConverter :303, UniformRandomSampler :376,
LinfMutation :399): genes live in the converter's scaled [0,1] feature
space as one dense matrix, scores as another, so survival and mutation
are plain NumPy ops. NSGA2Survival plugs `nsga2.nsga2_survival` into
the template's Survival slot.
"""

from __future__ import annotations

import dataclasses
import json
from typing import Optional, Sequence, Union

import numpy as np

from vizier_amd import pyvizier as vz
from vizier_amd.interfaces import serializable
from vizier_amd.converters.core import TrialToArrayConverter
from vizier_amd._src.algorithms.evolution import nsga2
from vizier_amd._src.algorithms.evolution import templates

_POPULATION_KEY = 'numpy_population'


@dataclasses.dataclass(frozen=True)
class Offspring:
  """Genes only — scores are attached once trials complete."""

  xs: np.ndarray  # (N, D) scaled features

  def __len__(self) -> int:
    return len(self.xs)


@dataclasses.dataclass(frozen=True)
class Population(templates.Population):
  """Evaluated genes: xs (N, D), ys (N, M) maximization scores, ids."""

  xs: np.ndarray
  ys: np.ndarray
  ids: np.ndarray  # (N,) originating trial ids (0 = unknown)

  def __len__(self) -> int:
    return len(self.xs)

  def __getitem__(self, index: Union[int, slice, np.ndarray]
                  ) -> 'Population':
    if isinstance(index, int):
      index = slice(index, index + 1)
    return Population(self.xs[index], self.ys[index], self.ids[index])

  def __add__(self, other: 'Population') -> 'Population':
    if not len(self):
      return other
    if not len(other):
      return self
    return Population(np.concatenate([self.xs, other.xs]),
                      np.concatenate([self.ys, other.ys]),
                      np.concatenate([self.ids, other.ids]))

  def dump(self) -> vz.Metadata:
    md = vz.Metadata()
    md[_POPULATION_KEY] = json.dumps({
        'xs': self.xs.tolist(), 'ys': self.ys.tolist(),
        'ids': self.ids.tolist()})
    return md

  @classmethod
  def recover(cls, metadata: vz.Metadata) -> 'Population':
    try:
      blob = metadata[_POPULATION_KEY]
    except KeyError as e:
      raise serializable.HarmlessDecodeError(
          'No population in metadata') from e
    try:
      d = json.loads(blob)
      return cls(np.asarray(d['xs'], dtype=np.float64).reshape(
                     len(d['xs']), -1),
                 np.asarray(d['ys'], dtype=np.float64).reshape(
                     len(d['ys']), -1),
                 np.asarray(d['ids'], dtype=np.int64))
    except (ValueError, KeyError, TypeError) as e:
      raise serializable.FatalDecodeError(
          f'Corrupt population blob: {e}') from e

  @classmethod
  def empty(cls, n_features: int, n_metrics: int) -> 'Population':
    return cls(np.zeros((0, n_features)), np.zeros((0, n_metrics)),
               np.zeros((0,), dtype=np.int64))


class PopulationConverter(templates.PopulationConverter[Population,
                                                        Offspring]):
  """Maps trials to scaled features / maximization labels."""

  def __init__(self, problem: vz.ProblemStatement):
    self._problem = problem
    self._converter = TrialToArrayConverter(problem)
    self._n_metrics = len(problem.metric_information)

  @property
  def n_features(self) -> int:
    return self._converter.n_features

  def to_population(self, completed: Sequence[vz.Trial]) -> Population:
    trials = [t for t in completed
              if t.final_measurement is not None and not t.infeasible]
    if not trials:
      return Population.empty(self._converter.n_features, self._n_metrics)
    xs = self._converter.to_features(trials).astype(np.float64)
    ys = self._converter.to_labels(trials).astype(np.float64)
    ids = np.asarray([t.id or 0 for t in trials], dtype=np.int64)
    keep = ~np.isnan(ys).any(axis=1)
    return Population(xs[keep], ys[keep], ids[keep])

  def to_suggestions(self, offsprings: Offspring
                     ) -> Sequence[vz.TrialSuggestion]:
    return [vz.TrialSuggestion(p)
            for p in self._converter.to_parameters(offsprings.xs)]


class UniformRandomSampler(templates.Sampler[Offspring]):

  def __init__(self, n_features: int, *, seed: Optional[int] = None):
    self._n_features = n_features
    self._rng = np.random.default_rng(seed)

  def sample(self, count: int) -> Offspring:
    return Offspring(self._rng.uniform(0, 1, (count, self._n_features)))


class LinfMutation(templates.Mutation[Population, Offspring]):
  """Perturb a random parent by at most `norm` per dimension."""

  def __init__(self, norm: float = 0.1, *, seed: Optional[int] = None):
    self._norm = norm
    self._rng = np.random.default_rng(seed)

  def mutate(self, population: Population, count: int) -> Offspring:
    parents = self._rng.integers(0, len(population), count)
    noise = self._rng.uniform(-self._norm, self._norm,
                              (count, population.xs.shape[1]))
    return Offspring(np.clip(population.xs[parents] + noise, 0.0, 1.0))


class NSGA2Survival(templates.Survival[Population]):
  """Pareto-rank + crowding-distance truncation (nsga2.py:57)."""

  def __init__(self, population_size: int):
    self._population_size = population_size

  def select(self, population: Population) -> Population:
    if len(population) <= self._population_size:
      return population
    idx = nsga2.nsga2_survival(population.ys, self._population_size)
    return population[idx]


def canonical_nsga2(problem: vz.ProblemStatement, *,
                    population_size: int = 50,
                    mutation_norm: float = 0.1,
                    first_survival_after: Optional[int] = None,
                    seed: Optional[int] = None,
                    ) -> templates.CanonicalEvolutionDesigner:
  """NSGA-II assembled from the template building blocks."""
  converter = PopulationConverter(problem)
  return templates.CanonicalEvolutionDesigner(
      converter,
      UniformRandomSampler(converter.n_features, seed=seed),
      NSGA2Survival(population_size),
      adaptation=LinfMutation(mutation_norm, seed=seed),
      first_survival_after=first_survival_after,
      population_size=population_size)
