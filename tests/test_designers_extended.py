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


class TestEagleUtils:
  """Behavior-level checks mirroring eagle_strategy_utils_test.py."""

  def _utils(self, problem=None, **cfg_kw):
    from vizier_amd._src.algorithms.designers.eagle_strategy.eagle_utils \
        import EagleUtils, FireflyAlgorithmConfig
    problem = problem or mixed_eagle_problem()
    return EagleUtils(problem, FireflyAlgorithmConfig(**cfg_kw),
                      np.random.default_rng(0))

  def test_pool_capacity_formula(self):
    # min(10 + round((df^1.2 + df) * 0.5), 1000) — reference :235.
    u = self._utils(continuous_problem(4))
    assert u.pool_capacity() == 10 + round((4 ** 1.2 + 4) * 0.5)
    u100 = self._utils(continuous_problem(100))
    expected = min(10 + round((100 ** 1.2 + 100) * 0.5), 1000)
    assert u100.pool_capacity() == expected

  def test_pull_weights_per_type_visibility(self):
    u = self._utils()
    v1 = {'f': 0.2, 'c': 'a', 'd': 0.0, 'i': 0.5}
    v2 = {'f': 0.8, 'c': 'b', 'd': 1.0, 'i': 0.0}
    w = u.pull_weights_by_type(v1, v2, other_is_better=True)
    # Continuous visibility (3.0) decays harder than categorical (0.2)
    # for the same per-dof distance scale.
    assert 0 < w[vz.ParameterType.DOUBLE] < 1
    assert 0 < w[vz.ParameterType.CATEGORICAL] <= 1
    # Worse fly pushes (negative weight scaled by negative_gravity).
    w_neg = u.pull_weights_by_type(v1, v2, other_is_better=False)
    assert w_neg[vz.ParameterType.DOUBLE] < 0

  def test_matching_categories_weaken_pull(self):
    # Reference counts EQUAL categories into distance^2 (:216).
    u = self._utils()
    same = {'f': 0.5, 'c': 'a', 'd': 0.5, 'i': 0.5}
    diff = {'f': 0.5, 'c': 'b', 'd': 0.5, 'i': 0.5}
    w_same = u.pull_weights_by_type(same, same, True)
    w_diff = u.pull_weights_by_type(diff, same, True)
    assert w_same[vz.ParameterType.CATEGORICAL] < \
        w_diff[vz.ParameterType.CATEGORICAL]

  def test_combine_categorical_bernoulli(self):
    u = self._utils()
    cfg = [c for c in u.parameter_configs if c.name == 'c'][0]
    # Extreme weights are deterministic.
    assert u.combine(cfg, 'a', 'b', 1.5) == 'a'
    assert u.combine(cfg, 'a', 'b', -0.5) == 'b'
    picks = [u.combine(cfg, 'a', 'b', 0.8) for _ in range(200)]
    frac_a = sum(p == 'a' for p in picks) / 200
    assert 0.65 < frac_a < 0.95

  def test_combine_numeric_is_linear_mix_clipped(self):
    u = self._utils()
    cfg = [c for c in u.parameter_configs if c.name == 'f'][0]
    assert u.combine(cfg, 0.8, 0.2, 0.5) == pytest.approx(0.5)
    # Weights outside [0,1] extrapolate, then clip to the scaled box.
    assert u.combine(cfg, 1.0, 0.0, 2.0) == 1.0

  def test_perturbation_scales_by_type(self):
    u = self._utils()
    scales = u.perturbation_scales()
    names = [c.name for c in u.parameter_configs]
    assert scales[names.index('c')] == 25.0      # categorical factor
    # Discrete: 10 / (n_feasible * base_perturbation).
    d_cfg = [c for c in u.parameter_configs if c.name == 'd'][0]
    nfeas = len(d_cfg.feasible_values)
    assert scales[names.index('d')] == pytest.approx(
        10.0 / (nfeas * 0.1))
    assert scales[names.index('f')] == 1.0

  def test_categorical_perturb_is_replacement_probability(self):
    u = self._utils()
    cfg = [c for c in u.parameter_configs if c.name == 'c'][0]
    kept = sum(u.perturb(cfg, 'a', 0.0) == 'a' for _ in range(50))
    assert kept == 50
    changed = sum(u.perturb(cfg, 'a', 1.0) != 'a' for _ in range(200))
    # With prob 1 a uniform category is drawn; 'a' itself has 1/3 mass.
    assert changed > 100

  def test_pure_categorical_constant_perturbation(self):
    p = vz.ProblemStatement()
    p.search_space.root.add_categorical_param('c1', ['a', 'b'])
    p.search_space.root.add_categorical_param('c2', ['x', 'y'])
    p.metric_information.append(vz.MetricInformation(
        name='obj', goal=vz.ObjectiveMetricGoal.MAXIMIZE))
    u = self._utils(p)
    assert u.is_pure_categorical()
    np.testing.assert_allclose(u.create_perturbations(0.3),
                               [0.1, 0.1])


def mixed_eagle_problem():
  p = vz.ProblemStatement()
  root = p.search_space.root
  root.add_float_param('f', 0.0, 10.0)
  root.add_categorical_param('c', ['a', 'b', 'z'])
  root.add_discrete_param('d', [1.0, 2.0, 4.0, 8.0])
  root.add_int_param('i', 0, 10)
  p.metric_information.append(vz.MetricInformation(
      name='obj', goal=vz.ObjectiveMetricGoal.MAXIMIZE))
  return p


class TestEagleDesignerBehavior:

  def _completed(self, designer, s, uid, value):
    t = s.to_trial(uid)
    t.complete(vz.Measurement(metrics={'obj': value}))
    designer.update(CompletedTrials([t]), ActiveTrials())
    return t

  def test_stuck_fly_escalates_perturbation(self):
    from vizier_amd._src.algorithms.designers.eagle_strategy.eagle_utils \
        import Firefly
    problem = continuous_problem(2)
    designer = EagleStrategyDesigner(problem, seed=0)
    # Install one fly and report a NON-improving trial with IDENTICAL
    # parameters: perturbation must x10 (capped at max_perturbation).
    values = {'x0': 0.5, 'x1': 0.5}
    designer._firefly_pool._pool[7] = Firefly(
        id_=7, perturbation=0.1, generation=1, values=dict(values),
        reward=1.0)
    s = vz.TrialSuggestion({'x0': 0.5, 'x1': 0.5})
    s.metadata.ns('eagle')['parent_fly_id'] = '7'
    self._completed(designer, s, 1, 0.0)   # worse reward
    assert designer._pool[7].perturbation == pytest.approx(0.5)

  def test_penalize_decay_and_best_fly_survives(self):
    from vizier_amd._src.algorithms.designers.eagle_strategy.eagle_utils \
        import Firefly
    problem = continuous_problem(2)
    designer = EagleStrategyDesigner(problem, seed=0)
    pool = designer._firefly_pool
    # Fill to capacity; fly 0 is the best with perturbation at floor.
    for i in range(pool.capacity):
      pool._pool[i] = Firefly(
          id_=i, perturbation=1e-3 if i == 0 else 0.1, generation=1,
          values={'x0': i / 100.0, 'x1': 0.5},
          reward=10.0 if i == 0 else float(i) / 100)
    s = vz.TrialSuggestion({'x0': 0.9, 'x1': 0.9})
    s.metadata.ns('eagle')['parent_fly_id'] = '0'
    self._completed(designer, s, 1, -1.0)  # non-improving
    # perturbation decayed below the bound, but the BEST fly survives.
    assert 0 in designer._pool
    # A non-best fly at the floor gets removed at capacity.
    designer._pool[1].perturbation = 1e-3
    s2 = vz.TrialSuggestion({'x0': 0.8, 'x1': 0.8})
    s2.metadata.ns('eagle')['parent_fly_id'] = '1'
    self._completed(designer, s2, 2, -1.0)
    assert 1 not in designer._pool

  def test_foreign_trial_adopted_by_closest_parent_only_if_better(self):
    from vizier_amd._src.algorithms.designers.eagle_strategy.eagle_utils \
        import Firefly
    problem = continuous_problem(2)
    designer = EagleStrategyDesigner(problem, seed=0)
    pool = designer._firefly_pool
    for i in range(pool.capacity):
      pool._pool[i] = Firefly(
          id_=i, perturbation=0.1, generation=1,
          values={'x0': i / pool.capacity, 'x1': 0.0}, reward=0.5)
    pool._max_fly_id = pool.capacity
    # A trial with NO eagle metadata lands near fly 0; better reward
    # replaces the closest fly's values.
    s = vz.TrialSuggestion({'x0': 0.01, 'x1': 0.02})
    self._completed(designer, s, 1, 2.0)
    assert designer._pool[0].reward == pytest.approx(2.0)

  def test_mixed_space_round_trip(self):
    problem = mixed_eagle_problem()
    designer = EagleStrategyDesigner(problem, seed=1)
    trials = RandomMetricsRunner(problem, iters=30, batch_size=1,
                                 seed=2).run_designer(designer)
    assert len(trials) == 30
    for t in trials:
      assert t.parameters.get_value('d') in (1.0, 2.0, 4.0, 8.0)
      assert t.parameters.get_value('c') in ('a', 'b', 'z')
      assert isinstance(t.parameters.get_value('i'), int)

  def test_serialization_roundtrip_preserves_rng_stream(self):
    problem = continuous_problem(3)
    d1 = EagleStrategyDesigner(problem, seed=9)
    trials = RandomMetricsRunner(problem, iters=25, batch_size=1,
                                 seed=3).run_designer(d1)
    assert trials
    state = d1.dump()
    d2 = EagleStrategyDesigner(problem, seed=123)  # different seed
    d2.load(state)
    s1 = [s.parameters.as_dict() for s in d1.suggest(3)]
    s2 = [s.parameters.as_dict() for s in d2.suggest(3)]
    assert s1 == s2


class TestBOCS:

  def test_runs_and_finds_good_bits(self):
    problem = bool_problem(5)
    designer = BOCSDesigner(problem, seed=7, sa_iters=100,
                            acquisition_optimizer='sa')

    def objective(s):
      bits = [s.parameters.get_value(f'b{i}') == 'true' for i in range(5)]
      return float(sum(bits[:3]) - sum(bits[3:]))

    trials = run_with_objective(designer, problem, objective, iters=30)
    best = max(t.final_measurement.metrics['obj'].value for t in trials)
    assert best >= 2.0

  def test_sdp_acquisition_default_finds_good_bits(self):
    # SDP relaxation + GW rounding is the reference DEFAULT acquisition
    # (bocs.py:537-539); solved here by Burer-Monteiro descent.
    problem = bool_problem(5)
    designer = BOCSDesigner(problem, seed=11)
    assert designer._acquisition_optimizer == 'sdp'

    def objective(s):
      bits = [s.parameters.get_value(f'b{i}') == 'true' for i in range(5)]
      return float(sum(bits[:3]) - sum(bits[3:]))

    trials = run_with_objective(designer, problem, objective, iters=30)
    best = max(t.final_measurement.metrics['obj'].value for t in trials)
    assert best >= 2.0

  def test_sdp_solves_known_quadratic(self):
    # Direct check of the SDP path: maximize a quadratic with known
    # argmax x = (1,1,0) via a hand-built weight vector.
    problem = bool_problem(3)
    designer = BOCSDesigner(problem, seed=3)
    # weights layout: [const, b1..b3, a_01, a_02, a_12] (maximize).
    weights = np.array([0.0, 2.0, 2.0, -1.0, 3.0, -2.0, -2.0])
    bits = designer._sdp_rounding(weights)
    np.testing.assert_array_equal(bits, [1.0, 1.0, 0.0])


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
