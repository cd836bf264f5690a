"""Tests for the evolution template framework and trial caches."""

import numpy as np
import pytest

from vizier_amd import pyvizier as vz
from vizier_amd.interfaces import serializable
from vizier_amd._src.algorithms.core.abstractions import (
    ActiveTrials,
    CompletedTrials,
)
from vizier_amd._src.algorithms.evolution import numpy_populations as npop
from vizier_amd._src.algorithms.evolution import templates
from vizier_amd._src.algorithms.policies.trial_caches import (
    IdDeduplicatingTrialLoader,
)
from vizier_amd._src.pythia.local_policy_supporters import (
    InRamPolicySupporter,
)


def _mo_problem():
  problem = vz.ProblemStatement()
  problem.search_space.root.add_float_param('x', 0.0, 1.0)
  problem.search_space.root.add_float_param('y', 0.0, 1.0)
  problem.metric_information.append(vz.MetricInformation(
      name='f0', goal=vz.ObjectiveMetricGoal.MAXIMIZE))
  problem.metric_information.append(vz.MetricInformation(
      name='f1', goal=vz.ObjectiveMetricGoal.MAXIMIZE))
  return problem


def _complete(suggestions, start_id=1):
  trials = []
  for i, s in enumerate(suggestions):
    t = s.to_trial(start_id + i)
    x = t.parameters.get_value('x')
    y = t.parameters.get_value('y')
    t.complete(vz.Measurement(metrics={'f0': x, 'f1': 1 - y}))
    trials.append(t)
  return trials


class TestPopulation:

  def test_sequence_semantics(self):
    p = npop.Population(np.arange(6.0).reshape(3, 2),
                        np.arange(6.0).reshape(3, 2) * 10,
                        np.array([1, 2, 3]))
    assert len(p) == 3
    assert len(p[0]) == 1            # int index keeps Population type
    assert isinstance(p[0], npop.Population)
    assert len(p[1:]) == 2
    combined = p + p[0:1]
    assert len(combined) == 4
    np.testing.assert_array_equal(combined.ids, [1, 2, 3, 1])

  def test_add_empty(self):
    p = npop.Population(np.ones((2, 3)), np.ones((2, 1)),
                        np.array([1, 2]))
    e = npop.Population.empty(3, 1)
    assert len(p + e) == 2 and len(e + p) == 2

  def test_dump_recover_roundtrip(self):
    p = npop.Population(np.random.default_rng(0).random((4, 2)),
                        np.random.default_rng(1).random((4, 2)),
                        np.array([3, 1, 4, 1]))
    q = npop.Population.recover(p.dump())
    np.testing.assert_allclose(q.xs, p.xs)
    np.testing.assert_allclose(q.ys, p.ys)
    np.testing.assert_array_equal(q.ids, p.ids)

  def test_recover_missing_raises_harmless(self):
    with pytest.raises(serializable.HarmlessDecodeError):
      npop.Population.recover(vz.Metadata())


class TestCanonicalEvolutionDesigner:

  def test_samples_then_mutates(self):
    problem = _mo_problem()
    designer = npop.canonical_nsga2(problem, population_size=8,
                                    first_survival_after=8, seed=0)
    # Before any updates: sampler path.
    first = designer.suggest(8)
    assert len(first) == 8
    trials = _complete(first)
    designer.update(CompletedTrials(trials), ActiveTrials())
    assert len(designer.population) == 8
    # After enough trials: mutation path still yields valid params.
    more = designer.suggest(5)
    assert len(more) == 5
    for s in more:
      assert 0.0 <= s.parameters.get_value('x') <= 1.0

  def test_survival_truncates_to_population_size(self):
    problem = _mo_problem()
    designer = npop.canonical_nsga2(problem, population_size=4,
                                    first_survival_after=4, seed=1)
    trials = _complete(designer.suggest(12))
    designer.update(CompletedTrials(trials), ActiveTrials())
    assert len(designer.population) == 4

  def test_survivors_prefer_pareto_front(self):
    problem = _mo_problem()
    designer = npop.canonical_nsga2(problem, population_size=2,
                                    first_survival_after=2, seed=2)
    # Two dominated points and two non-dominated ones.
    params = [(0.9, 0.1), (0.1, 0.9), (0.2, 0.8), (0.05, 0.95)]
    suggestions = [vz.TrialSuggestion({'x': x, 'y': y})
                   for x, y in params]
    trials = _complete(suggestions)
    designer.update(CompletedTrials(trials), ActiveTrials())
    # (0.9, f1=0.9) dominates everything with smaller x and smaller 1-y;
    # survivors must include the extreme point id 1.
    assert 1 in designer.population.ids

  def test_dump_load_roundtrip(self):
    problem = _mo_problem()
    designer = npop.canonical_nsga2(problem, population_size=4,
                                    first_survival_after=4, seed=3)
    trials = _complete(designer.suggest(6))
    designer.update(CompletedTrials(trials), ActiveTrials())
    md = designer.dump()
    fresh = npop.canonical_nsga2(problem, population_size=4,
                                 first_survival_after=4, seed=3)
    fresh.load(md)
    np.testing.assert_allclose(fresh.population.xs,
                               designer.population.xs)

  def test_adaptation_callable_selected_by_trials_seen(self):
    problem = _mo_problem()
    converter = npop.PopulationConverter(problem)
    calls = []

    class Recorder(templates.Mutation):
      def __init__(self, tag):
        self._tag = tag
      def mutate(self, population, count):
        calls.append(self._tag)
        return npop.Offspring(population.xs[:count])

    designer = templates.CanonicalEvolutionDesigner(
        converter,
        npop.UniformRandomSampler(converter.n_features, seed=0),
        npop.NSGA2Survival(4),
        adaptation=Recorder('default'),
        adaptation_callable=lambda n: Recorder(f'n={n}'),
        first_survival_after=2,
        population_size=4)
    trials = _complete(designer.suggest(4))
    designer.update(CompletedTrials(trials), ActiveTrials())
    designer.suggest(2)
    assert calls == ['n=4']


class TestIdDeduplicatingTrialLoader:

  def _setup(self):
    problem = _mo_problem()
    supporter = InRamPolicySupporter(problem)
    suggestions = [vz.TrialSuggestion({'x': 0.5, 'y': 0.5})
                   for _ in range(5)]
    trials = supporter.AddSuggestions(suggestions)
    for t in trials[:3]:
      t.complete(vz.Measurement(metrics={'f0': 1.0, 'f1': 1.0}))
    return supporter, trials

  def test_returns_each_completed_trial_once(self):
    supporter, trials = self._setup()
    loader = IdDeduplicatingTrialLoader(supporter)
    first = loader.get_newly_completed_trials(5)
    assert sorted(t.id for t in first) == [1, 2, 3]
    assert loader.get_newly_completed_trials(5) == []
    assert loader.num_incorporated_trials() == 3
    # A newly completed trial shows up on the next call.
    trials[3].complete(vz.Measurement(metrics={'f0': 0.0, 'f1': 0.0}))
    assert [t.id for t in loader.get_newly_completed_trials(5)] == [4]

  def test_active_trials_and_clear(self):
    supporter, _ = self._setup()
    loader = IdDeduplicatingTrialLoader(supporter)
    assert len(loader.get_active_trials()) == 2
    loader.get_newly_completed_trials(5)
    loader.clear()
    assert len(loader.get_newly_completed_trials(5)) == 3

  def test_dump_load_roundtrip(self):
    supporter, _ = self._setup()
    loader = IdDeduplicatingTrialLoader(supporter)
    loader.get_newly_completed_trials(5)
    md = loader.dump()
    fresh = IdDeduplicatingTrialLoader(supporter)
    fresh.load(md)
    assert fresh.num_incorporated_trials() == 3
    assert fresh.get_newly_completed_trials(5) == []

  def test_load_empty_metadata_resets(self):
    supporter, _ = self._setup()
    loader = IdDeduplicatingTrialLoader(supporter)
    loader.get_newly_completed_trials(5)
    loader.load(vz.Metadata())
    assert loader.num_incorporated_trials() == 0
