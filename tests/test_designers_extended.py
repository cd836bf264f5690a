"""Tests for NSGA2, CMA-ES, Eagle designer, BOCS, Harmonica."""

import numpy as np
import pytest

from vizier_amd import pyvizier as vz
from vizier_amd._src.algorithms.core.abstractions import (
    ActiveTrials,
    CompletedTrials,
)
from vizier_amd._src.algorithms.designers.bocs import BOCSDesigner
from vizier_amd._src.algorithms.designers.cmaes import CMAESDesigner
from vizier_amd._src.algorithms.designers.eagle_strategy.eagle_strategy import (
    EagleStrategyDesigner,
)
from vizier_amd._src.algorithms.designers.harmonica import HarmonicaDesigner
from vizier_amd._src.algorithms.evolution import nsga2
from vizier_amd._src.algorithms.testing.test_runners import (
    RandomMetricsRunner,
    run_with_objective,
)


def continuous_problem(dim=4, goal=vz.ObjectiveMetricGoal.MAXIMIZE):
  problem = vz.ProblemStatement()
  for i in range(dim):
    problem.search_space.root.add_float_param(f'x{i}', 0.0, 1.0)
  problem.metric_information.append(
      vz.MetricInformation(name='obj', goal=goal))
  return problem


def bool_problem(n=6):
  problem = vz.ProblemStatement()
  for i in range(n):
    problem.search_space.root.add_bool_param(f'b{i}')
  problem.metric_information.append(vz.MetricInformation(
      name='obj', goal=vz.ObjectiveMetricGoal.MAXIMIZE))
  return problem


def multiobjective_problem(dim=3):
  problem = continuous_problem(dim)
  problem.metric_information.append(vz.MetricInformation(
      name='obj2', goal=vz.ObjectiveMetricGoal.MAXIMIZE))
  return problem


class TestNSGA2:

  def test_pareto_rank_and_crowding(self):
    ys = np.array([[1.0, 0.0], [0.0, 1.0], [0.5, 0.5], [0.1, 0.1]])
    ranks = nsga2.pareto_rank(ys)
    assert list(ranks[:3]) == [0, 0, 0]
    assert ranks[3] == 1
    crowd = nsga2.crowding_distance(ys[:3])
    assert np.isinf(crowd[0]) and np.isinf(crowd[1])

  def test_survival_prefers_spread(self):
    ys = np.array([[1.0, 0.0], [0.0, 1.0], [0.55, 0.5], [0.5, 0.55],
                   [0.52, 0.52]])
    idx = nsga2.nsga2_survival(ys, 4)
    assert len(idx) == 4
    assert 0 in idx and 1 in idx  # extremes always survive

  def test_runs_on_random_metrics(self):
    problem = multiobjective_problem()
    designer = nsga2.NSGA2Designer(problem, seed=0)
    trials = RandomMetricsRunner(problem, iters=5, batch_size=4,
                                 seed=1).run_designer(designer)
    assert len(trials) == 20

  def test_advances_front(self):
    problem = multiobjective_problem(2)
    designer = nsga2.NSGA2Designer(problem, seed=2,
                                   population_size=20)

    def objectives(s):
      x = np.array([s.parameters.get_value(f'x{i}') for i in range(2)])
      return float(x[0]), float(1 - x[0] * x[1])

    uid = 0
    for _ in range(30):
      batch = designer.suggest(4)
      done = []
      for s in batch:
        uid += 1
        o1, o2 = objectives(s)
        t = s.to_trial(uid)
        t.complete(vz.Measurement(metrics={'obj': o1, 'obj2': o2}))
        done.append(t)
      designer.update(CompletedTrials(done), ActiveTrials())
    assert designer._pop_x is not None
    assert len(designer._pop_x) == 20


class TestCMAES:

  def test_rejects_non_double(self):
    problem = bool_problem()
    with pytest.raises(ValueError):
      CMAESDesigner(problem)

  def test_converges_on_sphere(self):
    problem = continuous_problem(4)
    designer = CMAESDesigner(problem, seed=3)

    def objective(s):
      x = np.array([s.parameters.get_value(f'x{i}') for i in range(4)])
      return float(-((x - 0.3) ** 2).sum())

    trials = run_with_objective(designer, problem, objective, iters=25,
                                batch_size=8)
    best = max(t.final_measurement.metrics['obj'].value for t in trials)
    assert best > -0.01
    assert designer._sigma < 0.3  # step size adapted down


class TestEagleDesigner:

  def test_runs_and_serializes(self):
    problem = continuous_problem(3)
    designer = EagleStrategyDesigner(problem, seed=4)
    trials = RandomMetricsRunner(problem, iters=8, batch_size=2,
                                 seed=5).run_designer(designer)
    assert len(trials) == 16
    state = designer.dump()
    fresh = EagleStrategyDesigner(problem, seed=4)
    fresh.load(state)
    assert len(fresh._pool) == len(designer._pool)
    assert fresh.suggest(1)

  def test_improves_over_random(self):
    problem = continuous_problem(3)
    designer = EagleStrategyDesigner(problem, seed=6)

    def objective(s):
      x = np.array([s.parameters.get_value(f'x{i}') for i in range(3)])
      return float(-((x - 0.5) ** 2).sum())

    trials = run_with_objective(designer, problem, objective, iters=150)
    best = max(t.final_measurement.metrics['obj'].value for t in trials)
    rng = np.random.default_rng(6)
    rand_best = max(float(-((rng.uniform(0, 1, 3) - 0.5) ** 2).sum())
                    for _ in range(150))
    assert best >= rand_best - 1e-6


class TestBOCS:

  def test_runs_and_finds_good_bits(self):
    problem = bool_problem(5)
    designer = BOCSDesigner(problem, seed=7, sa_iters=100)

    def objective(s):
      bits = [s.parameters.get_value(f'b{i}') == 'true' for i in range(5)]
      return float(sum(bits[:3]) - sum(bits[3:]))

    trials = run_with_objective(designer, problem, objective, iters=30)
    best = max(t.final_measurement.metrics['obj'].value for t in trials)
    assert best >= 2.0


class TestHarmonica:

  def test_requires_boolean_space(self):
    with pytest.raises(ValueError):
      HarmonicaDesigner(continuous_problem())

  def test_restricts_important_variables(self):
    problem = bool_problem(6)
    designer = HarmonicaDesigner(problem, seed=8, num_init_samples=15)

    def objective(s):
      b0 = 1.0 if s.parameters.get_value('b0') == 'true' else -1.0
      b1 = 1.0 if s.parameters.get_value('b1') == 'true' else -1.0
      return 3.0 * b0 * b1  # strong pairwise term

    trials = run_with_objective(designer, problem, objective, iters=40)
    best = max(t.final_measurement.metrics['obj'].value for t in trials)
    assert best == 3.0
    # After restriction, suggestions should mostly satisfy b0*b1 = +1.
    post = [designer.suggest(1)[0] for _ in range(10)]
    agree = sum((s.parameters.get_value('b0') ==
                 s.parameters.get_value('b1')) for s in post)
    assert agree >= 8
