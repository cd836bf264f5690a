"""Numerics tests for the GP core (CPU torch path = kernel oracle)."""

import math

import numpy as np
import pytest
import torch

from vizier_amd._src.gp import acquisitions as acq_lib
from vizier_amd._src.gp import gp_model, lbfgs, output_warpers
from vizier_amd._src.gp.matern import gram_matern52, matern52


class TestMatern:

  def test_gram_against_naive_loop(self):
    rng = np.random.default_rng(0)
    x = torch.tensor(rng.uniform(0, 1, (12, 4)), dtype=torch.float64)
    ls = torch.tensor(rng.uniform(0.3, 2.0, (4,)), dtype=torch.float64)
    amp = torch.tensor(1.7, dtype=torch.float64)
    K = gram_matern52(x, None, ls, amp)
    for i in range(12):
      for j in range(12):
        r = math.sqrt(float((((x[i] - x[j]) / ls) ** 2).sum()))
        sr = math.sqrt(5) * r
        expected = 1.7 ** 2 * (1 + sr + sr * sr / 3) * math.exp(-sr)
        assert float(K[i, j]) == pytest.approx(expected, rel=1e-9)

  def test_gram_psd_and_diag(self):
    rng = np.random.default_rng(1)
    x = torch.tensor(rng.uniform(0, 1, (50, 8)), dtype=torch.float64)
    ls = torch.full((8,), 0.5, dtype=torch.float64)
    K = gram_matern52(x, None, ls, torch.tensor(2.0, dtype=torch.float64))
    assert torch.allclose(torch.diagonal(K),
                          torch.full((50,), 4.0, dtype=torch.float64))
    eigs = torch.linalg.eigvalsh(K)
    assert float(eigs.min()) > -1e-8

  def test_cross_gram_shape(self):
    x1 = torch.rand(7, 3)
    x2 = torch.rand(9, 3)
    K = gram_matern52(x1, x2, torch.ones(3), torch.tensor(1.0))
    assert K.shape == (7, 9)


class TestLBFGS:

  def test_batched_quadratic(self):
    # NOTE: loss_fn must apply the SAME objective to every row (the
    # parallel line search evaluates an (S*R, P) stack).
    A = torch.tensor([[3.0, 1.0], [1.0, 2.0]])
    target = torch.tensor([1.0, -2.0])

    def loss(x):
      d = x - target
      return torch.einsum('ri,ij,rj->r', d, A, d)

    x0 = torch.tensor([[0.0, 0.0], [5.0, -5.0], [-3.0, 3.0]])
    x, f = lbfgs.minimize_batched(loss, x0, max_iters=50)
    assert torch.allclose(x, target.expand(3, 2), atol=1e-4)
    assert float(f.max()) < 1e-7

  def test_rosenbrock_batch(self):
    def loss(x):
      a, b = x[:, 0], x[:, 1]
      return (1 - a) ** 2 + 100 * (b - a * a) ** 2

    x0 = torch.tensor([[-1.0, 1.0], [0.0, 0.0], [2.0, 2.0]])
    x, f = lbfgs.minimize_batched(loss, x0, max_iters=300)
    assert float(f.min()) < 1e-4


class TestGPTraining:

  def _make_data(self, n=40, d=3, noise=0.01, seed=0):
    rng = np.random.default_rng(seed)
    x = rng.uniform(0, 1, (n, d))
    y = np.sin(3 * x[:, 0]) + 0.5 * x[:, 1] ** 2 + \
        noise * rng.standard_normal(n)
    return (torch.tensor(x, dtype=torch.float32),
            torch.tensor(y, dtype=torch.float32))

  def test_fit_reduces_nll_and_interpolates(self):
    x, y = self._make_data()
    post = gp_model.train_gp(x, y, num_restarts=3, max_iters=40, seed=1)
    mean, stddev = post.predict(x)
    # In-sample predictions close to targets, small stddev.
    assert float((mean - y).abs().mean()) < 0.1
    assert float(stddev.mean()) < 0.3

  def test_posterior_variance_grows_off_data(self):
    x, y = self._make_data()
    post = gp_model.train_gp(x, y, num_restarts=2, max_iters=30, seed=2)
    _, stddev_on = post.predict(x[:5])
    far = torch.full((1, 3), 5.0)
    _, stddev_far = post.predict(far)
    assert float(stddev_far[0]) > float(stddev_on.mean()) * 2

  def test_kinv_matches_cholesky_solve_path(self):
    x, y = self._make_data(n=25)
    post = gp_model.train_gp(x, y, num_restarts=2, max_iters=30, seed=3)
    xq = torch.rand(10, 3)
    mean_a, std_a = post.predict(xq)
    post_nokinv = gp_model.GPPosterior(
        x=post.x, params=post.params, L=post.L, alpha=post.alpha,
        K_inv=None, nll=post.nll)
    mean_b, std_b = post_nokinv.predict(xq)
    assert torch.allclose(mean_a, mean_b, atol=1e-4)
    # The GEMM quadform's fp32 cancellation error is bounded by the
    # posterior cache's noise floor (~6e-5*amp^2 in variance, see
    # train_gp) — a few percent of amp in stddev at worst.
    amp = float(post.params.amplitude)
    assert torch.allclose(std_a, std_b, atol=0.05 * amp)

  def test_cholesky_jitter_recovers_singular(self):
    K = torch.ones(5, 5)  # rank-1, singular
    L = gp_model.cholesky_with_jitter(K, torch.tensor(1.0))
    assert torch.isfinite(L).all()


class TestAcquisitions:

  def test_ei_positive_and_monotone_in_mean(self):
    ei = acq_lib.EI(best_value=0.0)
    mean = torch.tensor([-1.0, 0.0, 1.0])
    stddev = torch.ones(3)
    vals = ei(mean, stddev)
    assert (vals > 0).all()
    assert vals[0] < vals[1] < vals[2]

  def test_ucb_lcb(self):
    mean, stddev = torch.tensor([1.0]), torch.tensor([0.5])
    assert float(acq_lib.UCB(1.8)(mean, stddev)) == pytest.approx(1.9)
    assert float(acq_lib.LCB(1.8)(mean, stddev)) == pytest.approx(0.1)

  def test_trust_region_radius_and_penalty(self):
    trusted = torch.rand(10, 4) * 0.1
    tr = acq_lib.TrustRegion(trusted)
    expected = 0.2 + 0.3 * 10 / (5 * 5)
    assert tr.trust_radius == pytest.approx(expected)
    near = trusted[0] + 0.01
    far = torch.full((4,), 0.99)
    xs = torch.stack([near, far])
    scores = torch.tensor([1.0, 100.0])
    out = tr.apply(xs, scores)
    assert float(out[0]) == 1.0
    assert float(out[1]) < -1e4 + 1

  def test_trust_region_dof_counts_categorical_params_once(self):
    # Regression for ADVICE r1 (medium): one 20-value categorical must
    # contribute ONE dof (reference acquisitions.py:752-768), not 20.
    from vizier_amd import pyvizier as vz
    from vizier_amd.converters.core import TrialToArrayConverter
    problem = vz.ProblemStatement()
    problem.search_space.root.add_float_param('a', 0.0, 1.0)
    problem.search_space.root.add_float_param('b', 0.0, 1.0)
    problem.search_space.root.add_categorical_param(
        'c', [f'v{i}' for i in range(20)])
    problem.metric_information.append(vz.MetricInformation(
        name='obj', goal=vz.ObjectiveMetricGoal.MAXIMIZE))
    conv = TrialToArrayConverter(problem)
    assert conv.n_features == 22
    trusted = torch.rand(30, 22)
    tr = acq_lib.TrustRegion.for_converter(trusted, conv)
    # dof = 2 continuous + 1 categorical param = 3.
    expected = 0.2 + 0.3 * 30 / (5 * 4)
    assert tr.trust_radius == pytest.approx(expected)
    # One-hot columns excluded from the L-inf distance.
    xs = trusted[0].clone()
    xs[2:] = 1.0 - xs[2:]  # flip every one-hot column
    assert float(tr.min_linf_distance(xs[None])) == pytest.approx(0.0)

  def test_trust_region_excludes_wide_gap_discretes(self):
    from vizier_amd import pyvizier as vz
    from vizier_amd.converters.core import TrialToArrayConverter
    problem = vz.ProblemStatement()
    problem.search_space.root.add_float_param('a', 0.0, 1.0)
    # Two feasible values -> scaled gap 1.0 > min_radius: excluded.
    problem.search_space.root.add_discrete_param('d', [0.0, 100.0])
    # Eleven evenly spaced values -> gap 0.1 <= 0.2: included.
    problem.search_space.root.add_discrete_param(
        'e', [float(v) for v in range(0, 101, 10)])
    problem.metric_information.append(vz.MetricInformation(
        name='obj', goal=vz.ObjectiveMetricGoal.MAXIMIZE))
    conv = TrialToArrayConverter(problem)
    mask, n_cat = acq_lib.converter_trust_masks(conv)
    assert n_cat == 0
    assert mask == [False, True, False]
    trusted = torch.rand(10, 3)
    tr = acq_lib.TrustRegion.for_converter(trusted, conv)
    # dof = 2 (a and e; d excluded).
    expected = 0.2 + 0.3 * 10 / (5 * 3)
    assert tr.trust_radius == pytest.approx(expected)

  def test_qei_mc_sampling_covariance_is_correct(self):
    # Regression: the batched qEI sampler must draw with covariance
    # L L^T = cov (y = L eps). The transposed product L^T eps has both
    # wrong marginal variances and wrong correlations for q > 1 (a
    # real bug found in r2: einsum 'sbq,bqr' instead of 'sc,brc').
    torch.manual_seed(0)
    cov = torch.tensor([[[4.0, 1.9], [1.9, 1.0]]])   # corr ~ 0.95
    mean = torch.zeros(1, 2)
    eps = torch.randn(200000, 2)
    # Reconstruct the sampler's draws to check their statistics.
    L = torch.linalg.cholesky(cov + 1e-4 * cov.diagonal(
        dim1=-2, dim2=-1).mean() * torch.eye(2))
    y = torch.einsum('sc,brc->sbr', eps, L)[:, 0, :]
    emp = (y.T @ y) / y.shape[0]
    assert torch.allclose(emp, cov[0], atol=0.05), emp
    # And the score itself is the MC qEI of that joint distribution.
    scores = acq_lib.qei_mc_scores(mean, cov, eps, best_value=0.0)
    want = (y - 0.0).clamp_min(0).amax(-1).mean()
    assert float(scores[0]) == pytest.approx(float(want), rel=1e-4)

  def test_qei_mc_diagonal_fallback(self):
    # A non-PD covariance element must fall back to independent
    # marginals instead of NaNs.
    cov = torch.tensor([[[1.0, 2.0], [2.0, 1.0]]])   # indefinite
    mean = torch.zeros(1, 2)
    eps = torch.randn(512, 2)
    scores = acq_lib.qei_mc_scores(mean, cov, eps, best_value=0.0)
    assert torch.isfinite(scores).all()

  def test_hv_scalarization(self):
    s = acq_lib.create_hv_scalarization(100, 2, seed=0)
    ys = torch.tensor([[1.0, 1.0], [0.1, 0.1]])
    vals = s(ys).mean(dim=0)
    assert float(vals[0]) > float(vals[1])


class TestOutputWarpers:

  def test_default_pipeline_finite(self):
    labels = np.array([[1.0], [2.0], [np.nan], [100.0], [-50.0]])
    warper = output_warpers.create_default_warper()
    out = warper.warp(labels)
    assert np.isfinite(out).all()
    # Order of finite labels preserved.
    assert out[0, 0] < out[1, 0] < out[3, 0]
    # NaN (infeasible) mapped below everything finite.
    assert out[2, 0] < out[0, 0]

  def test_halfrank_preserves_good_half(self):
    labels = np.arange(10, dtype=np.float64)[:, None]
    warped = output_warpers.HalfRankComponent().warp(labels.copy())
    np.testing.assert_allclose(warped[5:, 0], labels[5:, 0])
    assert (np.diff(warped[:, 0]) > 0).all()

  def test_log_warper_roundtrip(self):
    labels = np.array([[0.5], [1.0], [7.0], [3.0]])
    w = output_warpers.LogWarperComponent()
    warped = w.warp(labels.copy())
    back = w.unwarp(warped)
    np.testing.assert_allclose(back, labels, rtol=1e-10)

  def test_zscore(self):
    labels = np.array([[1.0], [2.0], [3.0]])
    out = output_warpers.ZScoreLabels().warp(labels)
    assert abs(out.mean()) < 1e-9

  def test_detect_outliers_marks_only_outliers(self):
    rng = np.random.default_rng(0)
    labels = np.concatenate([rng.normal(5.0, 1.0, 40),
                             [-1e7, -1e76]])[:, None]
    out = output_warpers.DetectOutliers().warp(labels.copy())
    assert np.isnan(out[-1, 0]) and np.isnan(out[-2, 0])
    assert np.isfinite(out[:40]).all()
    np.testing.assert_allclose(out[:40], labels[:40])

  def test_warp_outliers_pipeline_finite(self):
    rng = np.random.default_rng(1)
    labels = np.concatenate([rng.normal(0.0, 2.0, 30),
                             [np.nan, -1e76]])[:, None]
    out = output_warpers.create_warp_outliers_warper().warp(
        labels.copy())
    assert out.shape == labels.shape
    assert np.isfinite(out).all()

  def test_transform_to_gaussian_monotone(self):
    labels = np.array([[0.1], [3.0], [1.5], [7.0]])
    out = output_warpers.TransformToGaussian().warp(labels.copy())
    order_in = np.argsort(labels[:, 0])
    order_out = np.argsort(out[:, 0])
    np.testing.assert_array_equal(order_in, order_out)
    # With use_rank the spacing is rank-based but the order holds too.
    out_r = output_warpers.TransformToGaussian(use_rank=True).warp(
        labels.copy())
    np.testing.assert_array_equal(np.argsort(out_r[:, 0]), order_in)

  def test_infeasible_warper_keeps_bad_below_feasible(self):
    # Regression for ADVICE r1 (high): the shift must apply to ALL
    # entries after NaN substitution, so infeasible trials stay the
    # WORST labels, not the best (reference output_warpers.py
    # InfeasibleWarperComponent).
    w = output_warpers.InfeasibleWarperComponent()
    labels = np.array([[100.0], [101.0], [102.0], [np.nan]])
    out = w.warp(labels.copy())
    # The substituted infeasible entry is strictly below every feasible.
    assert out[3, 0] < out[:3, 0].min()
    # Relative order of feasible entries preserved.
    assert out[0, 0] < out[1, 0] < out[2, 0]
    # unwarp is the exact inverse on the feasible entries.
    back = w.unwarp(out)
    np.testing.assert_allclose(back[:3], labels[:3], rtol=1e-12)

  def test_infeasible_warper_centering(self):
    # E[warp] over the substituted array is ~0 weighted by p_feasible
    # construction: mean of output == mean(sub) + shift.
    w = output_warpers.InfeasibleWarperComponent()
    labels = np.array([[1.0], [2.0], [np.nan], [3.0]])
    out = w.warp(labels.copy())
    rng_ = np.nanmax(labels) - np.nanmin(labels)
    warped_bad = np.nanmin(labels) - (0.5 * rng_ + 1)
    sub = np.array([1.0, 2.0, warped_bad, 3.0])
    np.testing.assert_allclose(out[:, 0], sub + w._shift, rtol=1e-12)

  def test_linear_output_warper_roundtrip(self):
    y = np.array([[1.0, 10.0], [3.0, 20.0], [2.0, 12.0]])
    w = output_warpers.LinearOutputWarper.from_obs(y)
    z = w.warp(y)
    assert z.min() >= -2.0 - 1e-12 and z.max() <= 2.0 + 1e-12
    np.testing.assert_allclose(w.unwarp(z), y, rtol=1e-12)
    yt = torch.tensor(y)
    wt = output_warpers.LinearOutputWarper.from_obs(yt)
    assert torch.allclose(wt.unwarp(wt.warp(yt)), yt)


class TestTransferLearning:

  def test_stacked_residual_gp_improves_with_prior(self):
    import torch
    from vizier_amd._src.gp import transfer_learning

    def f(x):
      return torch.sin(3 * x[:, 0]) + 0.5 * x[:, 1]

    g = torch.Generator().manual_seed(0)
    x_prior = torch.rand(60, 2, generator=g)
    y_prior = f(x_prior)
    x_cur = torch.rand(8, 2, generator=g)
    y_cur = f(x_cur)
    stacked = transfer_learning.train_stacked_gp(
        [(x_prior, y_prior), (x_cur, y_cur)], num_restarts=2,
        max_iters=20, seed=0)
    from vizier_amd._src.gp import gp_model
    single = gp_model.train_gp(x_cur, y_cur, num_restarts=2,
                               max_iters=20, seed=0)
    x_test = torch.rand(64, 2, generator=g)
    y_test = f(x_test)
    m_stacked, _ = stacked.predict(x_test)
    m_single, _ = single.predict(x_test)
    err_stacked = float((m_stacked - y_test).abs().mean())
    err_single = float((m_single - y_test).abs().mean())
    assert err_stacked < err_single

  def test_combine_predictions_inflates_base_uncertainty(self):
    import torch
    from vizier_amd._src.gp.transfer_learning import combine_predictions
    out = combine_predictions(
        torch.zeros(3), torch.full((3,), 0.1), torch.ones(3),
        torch.full((3,), 1.0), num_obs_base=10, num_obs_top=10)
    assert torch.allclose(out.mean, torch.ones(3))
    assert (out.stddev > 0.1).all()


class TestMES:

  def test_mes_prefers_uncertainty_near_incumbent(self):
    import torch
    from vizier_amd._src.gp import acquisitions as acq_lib
    mes = acq_lib.MaxValueEntropySearch(best_value=1.0,
                                        max_value_spread=1.0, seed=0)
    mean = torch.tensor([0.9, 0.9])
    stddev = torch.tensor([0.01, 0.5])
    vals = mes(mean, stddev)
    assert float(vals[1]) > float(vals[0])


class TestAdamOptimizer:
  """Adam ARD alternative (gp/adam.py, reference optax_wrappers)."""

  def test_minimizes_quadratic_batch(self):
    from vizier_amd._src.gp import adam
    target = torch.tensor([[1.0, -2.0, 0.5], [3.0, 0.0, -1.0]])

    def loss_fn(x):
      return ((x - target) ** 2).sum(-1)

    x0 = torch.zeros(2, 3)
    x_best, f_best = adam.minimize_adam(loss_fn, x0, epochs=400,
                                        learning_rate=0.05)
    assert float(f_best.max()) < 0.05
    assert torch.allclose(x_best, target, atol=0.15)

  def test_trains_gp_hyperparameters(self):
    from vizier_amd._src.gp import adam, gp_model
    g = torch.Generator().manual_seed(0)
    x = torch.rand(40, 3, generator=g)
    y = torch.sin(4 * x[:, 0]) + 0.05 * torch.randn(40, generator=g)

    def loss_fn(raw):
      return gp_model.negative_log_marginal_likelihood(raw, x, y)

    raw0 = gp_model._init_raw(4, 3, torch.Generator().manual_seed(1),
                              'cpu', torch.float32)
    f0 = loss_fn(raw0).min()
    x_best, f_best = adam.minimize_adam(loss_fn, raw0, epochs=150,
                                        learning_rate=0.05,
                                        normalize_by=40.0)
    assert float(f_best.min()) < float(f0) - 1.0
    # The optimized posterior interpolates reasonably.
    params = gp_model.GPParams.from_raw(x_best[int(f_best.argmin())])
    assert torch.isfinite(params.lengthscales).all()

  def test_handles_inf_rows(self):
    from vizier_amd._src.gp import adam

    def loss_fn(x):
      loss = (x ** 2).sum(-1)
      return torch.where(x[:, 0] > 10, torch.full_like(loss, float('inf')),
                         loss)

    x0 = torch.tensor([[0.5, 0.5], [20.0, 0.0]])
    x_best, f_best = adam.minimize_adam(loss_fn, x0, epochs=100)
    assert float(f_best[0]) < 0.05


class TestAnalyticNLLGradient:
  """gp_model.nll_value_and_grad vs autograd (the large-N fit path)."""

  @pytest.mark.parametrize('n,d,r', [(30, 2, 3), (60, 4, 5), (90, 7, 2)])
  def test_matches_autograd(self, n, d, r):
    g = torch.Generator().manual_seed(n + d)
    x = torch.rand(n, d, generator=g).double()
    y = (torch.sin(3 * x[:, 0]) +
         0.1 * torch.randn(n, generator=g).double())
    raw = (torch.randn(r, d + 3, generator=g) * 0.6).double()
    nll_a, grad_a = gp_model.nll_value_and_grad(raw, x, y)
    raw_t = raw.clone().requires_grad_(True)
    nll_t = gp_model.negative_log_marginal_likelihood(raw_t, x, y)
    grad_t, = torch.autograd.grad(nll_t.sum(), raw_t)
    assert torch.allclose(nll_a, nll_t.detach(), rtol=1e-10)
    assert torch.allclose(grad_a, grad_t, atol=1e-7, rtol=1e-6)

  def test_chol_hint_matches_no_hint(self):
    """The line-search ladder hands its (L, info) to the gradient eval
    (lbfgs ladder_fn); with the factors of the SAME raw values the
    result must be identical to factoring from scratch."""
    g = torch.Generator().manual_seed(5)
    x = torch.rand(40, 3, generator=g).double()
    y = torch.randn(40, generator=g).double()
    raw = torch.randn(3, 6, generator=g).double() * 0.5
    nll_ref, grad_ref = gp_model.nll_value_and_grad(raw, x, y)
    _, L, info = gp_model.nll_values_with_chol(raw, x, y)
    nll_h, grad_h = gp_model.nll_value_and_grad(
        raw, x, y, chol_hint=(L, info))
    # Not bitwise: the ladder builds K via gram_matern52 while the
    # gradient eval builds it from explicit distances, so L differs in
    # the last ulp. Equal to fp64 rounding.
    assert torch.allclose(nll_h, nll_ref, rtol=1e-12)
    assert torch.allclose(grad_h, grad_ref, rtol=1e-8, atol=1e-8)

  def test_ladder_fn_end_to_end_matches_plain(self):
    """train_gp's ladder-cache plumbing must not change the optimum:
    run minimize_batched with and without ladder_fn."""
    from vizier_amd._src.gp import lbfgs
    g = torch.Generator().manual_seed(7)
    x = torch.rand(30, 2, generator=g).double()
    y = torch.sin(4 * x[:, 0]).double()
    raw0 = torch.randn(2, 5, generator=g).double() * 0.4

    def loss_fn(raw):
      return gp_model.negative_log_marginal_likelihood(raw, x, y)

    def vag(raw, hint=None):
      return gp_model.nll_value_and_grad(raw, x, y, chol_hint=hint)

    def ladder(raw):
      nll_v, L, info = gp_model.nll_values_with_chol(raw, x, y)
      return nll_v, (L, info)

    best_a, f_a = lbfgs.minimize_batched(
        loss_fn, raw0.clone(), max_iters=15, value_and_grad_fn=vag)
    best_b, f_b = lbfgs.minimize_batched(
        loss_fn, raw0.clone(), max_iters=15, value_and_grad_fn=vag,
        ladder_fn=ladder)
    assert torch.allclose(f_a, f_b, rtol=1e-6)
    assert torch.allclose(best_a, best_b, rtol=1e-4, atol=1e-4)

  def test_float32_accuracy_vs_f64_truth(self):
    """fp32 analytic grads are as accurate as fp32 autograd grads
    (both measured against the f64 analytic ground truth)."""
    g = torch.Generator().manual_seed(0)
    x = torch.rand(50, 3, generator=g)
    y = torch.randn(50, generator=g)
    raw = torch.randn(4, 6, generator=g) * 0.5
    _, truth = gp_model.nll_value_and_grad(raw.double(), x.double(),
                                           y.double())
    _, grad_a = gp_model.nll_value_and_grad(raw, x, y)
    raw_t = raw.clone().requires_grad_(True)
    nll_t = gp_model.negative_log_marginal_likelihood(raw_t, x, y)
    grad_t, = torch.autograd.grad(nll_t.sum(), raw_t)
    err_analytic = float((grad_a.double() - truth).abs().max())
    err_autograd = float((grad_t.double() - truth).abs().max())
    scale = float(truth.abs().max())
    assert err_analytic < max(3 * err_autograd, 1e-4 * scale), (
        err_analytic, err_autograd)

  def test_large_n_path_end_to_end(self, monkeypatch):
    """Full train_gp through the analytic-gradient + blocked-solve
    path (threshold lowered so it runs at CPU scale)."""
    monkeypatch.setattr(gp_model, '_NO_GRAD_FIT_N', 50)
    g = torch.Generator().manual_seed(1)
    x = torch.rand(80, 3, generator=g)
    y = torch.sin(4 * x[:, 0]) + 0.05 * torch.randn(80, generator=g)
    post = gp_model.train_gp(x, y, num_restarts=2, max_iters=30, seed=0)
    mean, stddev = post.predict(x)
    assert float((mean - y).abs().mean()) < 0.15
    assert (stddev > 0).all()
    # And it actually fitted (not the frozen default hyperparameters).
    assert post.nll < gp_model.negative_log_marginal_likelihood(
        gp_model._init_raw(1, 3, torch.Generator().manual_seed(0),
                           'cpu', torch.float32), x, y)[0] - 1.0


class TestEnsemblePosterior:

  def _fit(self, ensemble_size):
    g = torch.Generator().manual_seed(0)
    x = torch.rand(30, 2, generator=g)
    y = torch.randn(30, generator=g)
    return gp_model.train_gp(x, y, num_restarts=4, max_iters=10,
                             ensemble_size=ensemble_size), x

  def test_default_returns_single_posterior(self):
    post, _ = self._fit(1)
    assert isinstance(post, gp_model.GPPosterior)

  def test_ensemble_members_sorted_by_nll(self):
    post, _ = self._fit(3)
    assert isinstance(post, gp_model.EnsembleGPPosterior)
    nlls = [m.nll for m in post.members]
    assert nlls == sorted(nlls)
    # Duck-type surface used by the designers.
    assert post.params is post.members[0].params
    assert post.K_inv is None

  def test_mixture_moments_match_hand_computation(self):
    post, x = self._fit(3)
    g = torch.Generator().manual_seed(1)
    xq = torch.rand(7, 2, generator=g)
    mean, stddev = post.predict(xq)
    ms = torch.stack([m.predict(xq)[0] for m in post.members])
    ss = torch.stack([m.predict(xq)[1] for m in post.members])
    want_mean = ms.mean(0)
    want_var = (ss.square() + ms.square()).mean(0) - want_mean.square()
    assert torch.allclose(mean, want_mean, atol=1e-6)
    assert torch.allclose(stddev, want_var.clamp_min(1e-12).sqrt(),
                          atol=1e-6)
    # Mixture stddev >= smallest member stddev (spread adds variance).
    assert (stddev >= ss.min(0).values - 1e-6).all()

  def test_scoring_function_uses_mixture(self):
    from vizier_amd._src.gp import acquisitions as acq_lib
    post, _ = self._fit(3)
    g = torch.Generator().manual_seed(2)
    xq = torch.rand(5, 2, generator=g)
    scoring = acq_lib.ScoringFunction(post, acq_lib.UCB(1.8))
    mean, stddev = post.predict(xq)
    assert torch.allclose(scoring(xq), mean + 1.8 * stddev, atol=1e-6)
