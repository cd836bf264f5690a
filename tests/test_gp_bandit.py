"""GP-Bandit designer end-to-end (CPU; GPU variants in test_gpu_*)."""

import numpy as np
import pytest
import torch

from vizier_amd import pyvizier as vz
from vizier_amd._src.algorithms.core.abstractions import (
    ActiveTrials,
    CompletedTrials,
)
from vizier_amd._src.algorithms.designers.gp_bandit import (
    GPBanditConfig,
    VizierGPBandit,
)


def make_problem(dim=4) -> vz.ProblemStatement:
  problem = vz.ProblemStatement()
  for i in range(dim):
    problem.search_space.root.add_float_param(f'x{i}', 0.0, 1.0)
  problem.metric_information.append(
      vz.MetricInformation(name='obj', goal=vz.ObjectiveMetricGoal.MAXIMIZE))
  return problem


def evaluate(suggestion, optimum=0.7) -> float:
  x = np.array([suggestion.parameters.get_value(f'x{i}')
                for i in range(4)])
  return float(-((x - optimum) ** 2).sum())


def small_config(**kw) -> GPBanditConfig:
  defaults = dict(max_evaluations=1500, suggestion_batch_size=25,
                  ard_restarts=2, ard_max_iters=20, device='cpu')
  defaults.update(kw)
  return GPBanditConfig(**defaults)


def run_loop(designer, n_iters, batch=1):
  trials = []
  uid = 0
  for _ in range(n_iters):
    suggestions = designer.suggest(batch)
    assert suggestions, 'designer returned no suggestions'
    completed = []
    for s in suggestions:
      uid += 1
      t = s.to_trial(uid)
      t.complete(vz.Measurement(metrics={'obj': evaluate(s)}))
      completed.append(t)
    designer.update(CompletedTrials(completed), ActiveTrials())
    trials.extend(completed)
  return trials


class TestGPBandit:

  def test_seed_phase_center_first(self):
    designer = VizierGPBandit(make_problem(), small_config())
    first = designer.suggest(1)
    assert first[0].parameters.get_value('x0') == pytest.approx(0.5)

  def test_loop_converges_toward_optimum(self):
    designer = VizierGPBandit(make_problem(), small_config(), seed=1)
    trials = run_loop(designer, 12)
    values = [t.final_measurement.metrics['obj'].value for t in trials]
    best = max(values)
    # Random search on 4-D needs far more than 12 points to hit -0.05.
    assert best > -0.05, f'best={best}, values={values}'

  def test_gp_beats_random_search_same_budget(self):
    n_iters = 12
    designer = VizierGPBandit(make_problem(), small_config(), seed=3)
    gp_trials = run_loop(designer, n_iters)
    gp_best = max(t.final_measurement.metrics['obj'].value
                  for t in gp_trials)
    rng = np.random.default_rng(3)
    random_best = max(
        float(-((rng.uniform(0, 1, 4) - 0.7) ** 2).sum())
        for _ in range(n_iters))
    assert gp_best > random_best

  def test_mixed_space_suggestions_feasible(self):
    problem = vz.ProblemStatement()
    root = problem.search_space.root
    root.add_float_param('x', 0.0, 1.0)
    root.add_int_param('i', 1, 5)
    root.add_categorical_param('c', ['a', 'b', 'c'])
    root.add_discrete_param('d', [0.5, 1.5, 3.5])
    problem.metric_information.append(
        vz.MetricInformation(name='obj',
                             goal=vz.ObjectiveMetricGoal.MAXIMIZE))
    designer = VizierGPBandit(problem, small_config(max_evaluations=500),
                              seed=0)
    uid = 0
    for _ in range(5):
      for s in designer.suggest(1):
        for pc in problem.search_space.parameters:
          assert pc.contains(s.parameters.get_value(pc.name))
        uid += 1
        t = s.to_trial(uid)
        t.complete(vz.Measurement(
            metrics={'obj': float(s.parameters.get_value('x'))}))
        designer.update(CompletedTrials([t]), ActiveTrials())

  def test_qei_parallel_suggestions(self):
    designer = VizierGPBandit(
        make_problem(), small_config(acquisition='qei',
                                     max_evaluations=500), seed=2)
    run_loop(designer, 2)  # seed phase
    batch = designer.suggest(4)
    assert len(batch) == 4
    # The q-EI batch should be diverse, not 4 copies of one point.
    points = [tuple(s.parameters.as_dict().values()) for s in batch]
    assert len(set(points)) > 1

  def test_multi_objective_hv_scalarization(self):
    problem = make_problem()
    problem.metric_information.append(
        vz.MetricInformation(name='obj2',
                             goal=vz.ObjectiveMetricGoal.MINIMIZE))
    designer = VizierGPBandit(problem, small_config(max_evaluations=500),
                              seed=4)
    uid = 0
    for _ in range(5):
      for s in designer.suggest(1):
        uid += 1
        t = s.to_trial(uid)
        x0 = float(s.parameters.get_value('x0'))
        t.complete(vz.Measurement(metrics={'obj': x0, 'obj2': 1 - x0}))
        designer.update(CompletedTrials([t]), ActiveTrials())

  def test_predictor_interface(self):
    designer = VizierGPBandit(make_problem(), small_config(), seed=5)
    run_loop(designer, 4)
    pred = designer.predict([vz.TrialSuggestion(
        {f'x{i}': 0.5 for i in range(4)})])
    assert pred.mean.shape == (1, 1)
    assert pred.stddev.shape == (1, 1)
    assert np.isfinite(pred.mean).all()

  def test_infeasible_trials_handled(self):
    designer = VizierGPBandit(make_problem(), small_config(), seed=6)
    trials = run_loop(designer, 3)
    # Add an infeasible trial; next suggest should not crash.
    bad = designer.suggest(1)[0].to_trial(100)
    bad.complete(vz.Measurement(), infeasibility_reason='nan')
    designer.update(CompletedTrials([bad]), ActiveTrials())
    out = designer.suggest(1)
    assert out

  def test_transfer_learning_priors(self):
    """set_priors stacks a prior-study GP under the current study's GP
    (reference gp_bandit.py:289, gp/gp_models.py:245-365)."""
    problem = make_problem()
    rng = np.random.default_rng(5)
    prior_trials = []
    for uid in range(1, 41):
      x = rng.uniform(0, 1, 4)
      t = vz.Trial({f'x{i}': float(x[i]) for i in range(4)}, id=uid)
      t.complete(vz.Measurement(
          metrics={'obj': float(-((x - 0.7) ** 2).sum())}))
      prior_trials.append(t)

    designer = VizierGPBandit(problem, small_config(), seed=0)
    designer.set_priors([CompletedTrials(prior_trials)])
    assert designer._prior_stack is not None
    run_loop(designer, 5)
    # Residual stacking active: scoring + prediction go through the
    # stacked GP chain.
    assert designer._stacked is not None
    pred = designer.predict(
        [vz.TrialSuggestion({f'x{i}': 0.7 for i in range(4)})])
    assert pred.mean.shape == (1, 1) and np.isfinite(pred.mean).all()
    assert (pred.stddev > 0).all()

  def test_transfer_learning_resilient_to_bad_prior(self):
    """A misleading prior must not break convergence (the top GP learns
    the residual; reference 'resilient to bad priors')."""
    problem = make_problem()
    rng = np.random.default_rng(9)
    bad_prior = []
    for uid in range(1, 31):
      x = rng.uniform(0, 1, 4)
      # Prior points AWAY from the true optimum (optimum at 0.0).
      t = vz.Trial({f'x{i}': float(x[i]) for i in range(4)}, id=uid)
      t.complete(vz.Measurement(
          metrics={'obj': float(-((x - 0.0) ** 2).sum())}))
      bad_prior.append(t)

    designer = VizierGPBandit(problem, small_config(), seed=1)
    designer.set_priors([CompletedTrials(bad_prior)])
    trials = run_loop(designer, 8)
    best = max(t.final_measurement.metrics['obj'].value for t in trials)
    assert best > -0.4, f'transfer designer failed to adapt: {best}'


class TestEnsembleDesigners:

  def _problem(self):
    p = vz.ProblemStatement()
    for i in range(3):
      p.search_space.root.add_float_param(f'x{i}', -1.0, 1.0)
    p.metric_information.append(vz.MetricInformation(
        name='m', goal=vz.ObjectiveMetricGoal.MAXIMIZE))
    return p

  def _trials(self, n=14):
    rng = np.random.default_rng(0)
    trials = []
    for uid in range(1, n + 1):
      params = {f'x{i}': float(rng.uniform(-1, 1)) for i in range(3)}
      t = vz.Trial(params, id=uid)
      x = np.array(list(params.values()))
      t.complete(vz.Measurement(metrics={'m': float(-(x ** 2).sum())}))
      trials.append(t)
    return trials

  def test_gp_bandit_ensemble_end_to_end(self):
    from vizier_amd._src.gp.gp_model import EnsembleGPPosterior
    d = VizierGPBandit(self._problem(), GPBanditConfig(
        max_evaluations=400, ard_restarts=3, ard_max_iters=8,
        ensemble_size=3))
    d.update(CompletedTrials(self._trials()), ActiveTrials())
    assert len(d.suggest(1)) == 1
    assert isinstance(d._posteriors[0], EnsembleGPPosterior)
    # Warm refit keeps the ensemble and still suggests.
    t = vz.Trial({f'x{i}': 0.1 for i in range(3)}, id=99)
    t.complete(vz.Measurement(metrics={'m': 0.0}))
    d.update(CompletedTrials([t]), ActiveTrials())
    assert len(d.suggest(1)) == 1

  def test_gp_ucb_pe_ensemble_both_phases(self):
    from vizier_amd._src.algorithms.designers.gp_ucb_pe import (
        UCBPEConfig, VizierGPUCBPEBandit)
    from vizier_amd._src.gp.gp_model import EnsembleGPPosterior
    d = VizierGPUCBPEBandit(self._problem(), UCBPEConfig(
        max_evaluations=400, ard_restarts=3, ard_max_iters=8,
        ensemble_size=3))
    d.update(CompletedTrials(self._trials()), ActiveTrials())
    # count=3 exercises UCB (first) and PE fill (rest) with the mixture.
    assert len(d.suggest(3)) == 3
    assert isinstance(d._posterior, EnsembleGPPosterior)

  def test_custom_output_warper_and_ref_scaling(self):
    from vizier_amd._src.gp import output_warpers as ow
    d = VizierGPBandit(self._problem(), GPBanditConfig(
        max_evaluations=200, ard_restarts=2, ard_max_iters=5,
        output_warper_factory=ow.create_warp_outliers_warper,
        ref_scaling=0.1))
    d.update(CompletedTrials(self._trials(10)), ActiveTrials())
    assert len(d.suggest(1)) == 1

  def test_linear_coef_end_to_end_and_extrapolation(self):
    from vizier_amd._src.gp import linear_matern as lm
    import torch
    # The combined kernel captures a global linear trend.
    g = torch.Generator().manual_seed(0)
    x = torch.rand(40, 3, generator=g)
    y = 3.0 * x.sum(-1) + 0.3 * torch.sin(8 * x[:, 0])
    post = lm.train_linear_matern_gp(x, y, linear_coef=1.0,
                                     num_restarts=3, max_iters=20)
    xq = torch.rand(8, 3, generator=g)
    yq = 3.0 * xq.sum(-1) + 0.3 * torch.sin(8 * xq[:, 0])
    mean, stddev = post.predict(xq)
    assert float((mean - yq).abs().mean()) < 0.5
    assert (stddev > 0).all()
    # Outside the data the linear term keeps the trend direction
    # (a stationary kernel reverts to the constant mean ~2.2).
    m_far, _ = post.predict(torch.full((1, 3), 2.0))
    assert float(m_far) > float(y.mean()) + 1.0

    d = VizierGPBandit(self._problem(), GPBanditConfig(
        max_evaluations=300, ard_restarts=2, ard_max_iters=8,
        linear_coef=1.0))
    d.update(CompletedTrials(self._trials()), ActiveTrials())
    assert len(d.suggest(1)) == 1
    assert isinstance(d._posteriors[0], lm.LinearMaternPosterior)
    # qEI with the combined kernel (batched joint covariance path).
    d2 = VizierGPBandit(self._problem(), GPBanditConfig(
        max_evaluations=300, ard_restarts=2, ard_max_iters=8,
        acquisition='qei', linear_coef=1.0))
    d2.update(CompletedTrials(self._trials()), ActiveTrials())
    assert len(d2.suggest(4)) == 4

  def test_custom_scoring_function_factory(self):
    import torch
    calls = []

    def factory(posterior, best_value, trust_region):
      def score(xs: torch.Tensor) -> torch.Tensor:
        calls.append(xs.shape)
        mean, stddev = posterior.predict(xs)
        return mean + 0.5 * stddev  # custom low-explore UCB
      return score

    d = VizierGPBandit(self._problem(), GPBanditConfig(
        max_evaluations=200, ard_restarts=2, ard_max_iters=5,
        scoring_function_factory=factory))
    d.update(CompletedTrials(self._trials(10)), ActiveTrials())
    assert len(d.suggest(1)) == 1
    assert calls, 'custom scorer was never invoked'

  def test_default_config_keeps_fused_scorer_attach(self):
    # The GPU sweep (hipGraph + megakernel) keys off score_fn.scoring /
    # codec_identity; optional surrogates must not cost the default
    # path its fast-path eligibility.
    d = VizierGPBandit(self._problem(), GPBanditConfig(
        max_evaluations=200, ard_restarts=2, ard_max_iters=5))
    d.update(CompletedTrials(self._trials(10)), ActiveTrials())
    d._fit()
    score_fn, _ = d._score_factory(1)
    assert hasattr(score_fn, 'scoring')
    assert hasattr(score_fn, 'codec_identity')
    assert not getattr(score_fn, 'graph_safe', True) is False
