"""GP-UCB-PE designer: the DEFAULT algorithm.

Capability parity with vizier/_src/algorithms/designers/gp_ucb_pe.py
(VizierGPUCBPEBandit :609, UCBPEConfig :80, UCBScoreFunction :282,
PEScoreFunction :384, suggest flow :1356-1445): the first suggestion of
a batch exploits with UCB (coef 1.8); the remaining batch members run
Pure Exploration — maximize the predictive stddev conditioned on
completed AND pending/hallucinated points, linearly penalized (coef 10)
for leaving the promising region {x : UCB_explore(x) >= threshold},
where threshold is the predicted mean at the observed point with the
highest UCB. Probabilistic UCB/PE overwrites and the high-noise
exploration boost mirror the reference's UCBPEConfig defaults.
"""

from __future__ import annotations

import copy
import dataclasses
from typing import List, Optional, Sequence

import numpy as np
import torch

from vizier_amd import pyvizier as vz
from vizier_amd.converters.core import TrialToArrayConverter
from vizier_amd._src.algorithms.core.abstractions import (
    ActiveTrials,
    CompletedTrials,
    Designer,
)
from vizier_amd._src.algorithms.designers import gp_bandit
from vizier_amd._src.algorithms.optimizers.eagle import (
    CandidateBatch,
    EagleStrategyConfig,
)
from vizier_amd._src.algorithms.optimizers.vectorized import (
    EagleFeatureCodec,
    VectorizedOptimizerFactory,
    trials_to_sorted_features,
)
from vizier_amd._src.gp import acquisitions as acq_lib
from vizier_amd._src.gp import gp_model, output_warpers


@dataclasses.dataclass
class UCBPEConfig:
  """Reference defaults (gp_ucb_pe.py:80-130)."""

  ucb_coefficient: float = 1.8
  explore_region_ucb_coefficient: float = 0.5
  cb_violation_penalty_coefficient: float = 10.0
  ucb_overwrite_probability: float = 0.25
  pe_overwrite_probability: float = 0.1
  pe_overwrite_probability_in_high_noise: float = 0.7
  ensemble_size: int = 1   # best-N ARD restarts mixed (gp_ucb_pe.py:651)
  # Matérn + continuous-only linear kernel (gp_ucb_pe.py:676
  # _mixes_linear_kernel -> linear_coef=1.0). Composed scoring path.
  mixes_linear_kernel: bool = False
  num_scalarizations: int = 1000  # multimetric UCB scalarization dirs
  # Multimetric surrogate (gp_ucb_pe.py:129 multitask_type):
  # 'independent' per-metric GPs or 'separable'/'separable_diag' joint.
  multitask_type: str = 'independent'
  signal_to_noise_threshold: float = 0.7
  # Batch PE via the SET acquisition (gp_ucb_pe.py:118,510): the
  # exploration members of a batch are optimized JOINTLY, scored by the
  # log-determinant of their predictive covariance (conditioned on
  # completed + pending trials) — intra-batch correlation is penalized,
  # so the batch decorrelates, unlike the sequential summed-stddev fill.
  optimize_set_acquisition_for_exploration: bool = False
  max_evaluations: int = 75000
  suggestion_batch_size: int = 25
  num_seed_trials: int = 2
  ard_restarts: int = 4
  ard_max_iters: int = 50
  ard_warm_iters: int = 12
  ard_warm_restarts: int = 2
  use_trust_region: bool = True
  device: Optional[str] = None
  dtype: torch.dtype = torch.float32


class VizierGPUCBPEBandit(Designer):
  """UCB exploitation + Pure-Exploration batch fill."""

  def __init__(self, problem: vz.ProblemStatement,
               config: Optional[UCBPEConfig] = None, *, seed: int = 0):
    self._problem = problem
    self._config = config or UCBPEConfig()
    self._seed = seed
    self._rng = np.random.default_rng(seed)
    self._completed: List[vz.Trial] = []
    self._active: List[vz.Trial] = []
    self._converter = TrialToArrayConverter(problem)
    self._codec = EagleFeatureCodec(self._converter)
    self._device = self._config.device or gp_bandit.default_device()
    self._posterior: Optional[gp_model.GPPosterior] = None
    self._mo_posteriors = None    # per-metric GPs (multimetric studies)
    self._mo_scalarizer = None
    self._mt_raw = None           # joint multitask warm start
    self._last_fit_count = -1
    self._last_suggest_completed = 0
    from vizier_amd._src.algorithms.designers.quasi_random import (
        QuasiRandomDesigner,
    )
    self._quasi_random = (
        QuasiRandomDesigner(problem.search_space, seed=seed)
        if not problem.search_space.is_conditional else None)

  @classmethod
  def from_problem(cls, problem: vz.ProblemStatement,
                   seed: int = 0) -> 'VizierGPUCBPEBandit':
    return cls(problem, seed=seed)

  # -- Designer API ---------------------------------------------------------

  def update(self, completed: CompletedTrials, all_active: ActiveTrials
             ) -> None:
    self._completed.extend(completed.trials)
    self._active = list(all_active.trials)

  def suggest(self, count: Optional[int] = None
              ) -> Sequence[vz.TrialSuggestion]:
    count = count or 1
    cfg = self._config
    if len(self._completed) < cfg.num_seed_trials:
      return self._seed_suggestions(count)

    self._fit()
    has_new = len(self._completed) > self._last_suggest_completed
    self._last_suggest_completed = len(self._completed)

    # Hallucinated feature set starts as completed + currently-active.
    x_all = [self._posterior.x]
    if self._active:
      x_all.append(torch.as_tensor(
          self._converter.to_features(self._active), dtype=cfg.dtype,
          device=self._device))

    suggestions: List[vz.TrialSuggestion] = []
    if (cfg.optimize_set_acquisition_for_exploration and count > 1 and
        self._mo_posteriors is None):
      # Set-PE batch (gp_ucb_pe.py:1423-1437): at most ONE UCB exploit
      # suggestion (only when new completed trials arrived), then the
      # rest jointly via the logdet set acquisition.
      if has_new:
        x_cat = x_all[0] if len(x_all) == 1 else torch.cat(x_all, dim=0)
        dense = self._optimize_one(True, x_cat)
        x_all.append(dense.reshape(1, -1))
        params = self._converter.to_parameters(
            dense.detach().cpu().numpy())[0]
        s = vz.TrialSuggestion(params)
        s.metadata.ns('gp_ucb_pe')['acquisition'] = 'ucb'
        suggestions.append(s)
      remaining = count - len(suggestions)
      if remaining > 0:
        x_cat = x_all[0] if len(x_all) == 1 else torch.cat(x_all, dim=0)
        rows = self._optimize_set(x_cat, remaining)
        for p in self._converter.to_parameters(
            rows.detach().cpu().numpy()):
          s = vz.TrialSuggestion(p)
          s.metadata.ns('gp_ucb_pe')['acquisition'] = 'set_pe'
          suggestions.append(s)
      return suggestions

    for i in range(count):
      use_ucb = self._choose_ucb(has_new and i == 0 and not self._active)
      # Single-element cat would COPY, breaking the trust-region anchor
      # (posterior.x identity) that unlocks the fused scorer path.
      x_cat = x_all[0] if len(x_all) == 1 else torch.cat(x_all, dim=0)
      dense = self._optimize_one(use_ucb, x_cat)
      x_all.append(dense.reshape(1, -1))
      params = self._converter.to_parameters(
          dense.detach().cpu().numpy())[0]
      s = vz.TrialSuggestion(params)
      s.metadata.ns('gp_ucb_pe')['acquisition'] = \
          'ucb' if use_ucb else 'pe'
      suggestions.append(s)
    return suggestions

  # -- internals ------------------------------------------------------------

  def _seed_suggestions(self, count: int) -> List[vz.TrialSuggestion]:
    from vizier_amd._src.pythia import suggest_default
    out: List[vz.TrialSuggestion] = []
    if not self._completed and not self._active:
      out.append(vz.TrialSuggestion(suggest_default.get_default_parameters(
          self._problem.search_space)))
    remaining = count - len(out)
    if remaining > 0:
      if self._quasi_random is not None:
        out.extend(self._quasi_random.suggest(remaining))
      else:
        from vizier_amd._src.algorithms.designers.random import (
            RandomDesigner,
        )
        out.extend(RandomDesigner(self._problem.search_space,
                                  seed=self._seed).suggest(remaining))
    return out

  def _fit(self) -> None:
    if self._last_fit_count == len(self._completed) and self._posterior:
      return
    cfg = self._config
    x_np, y_np = self._converter.to_xy(self._completed)
    x = torch.as_tensor(x_np, dtype=cfg.dtype, device=self._device)

    if y_np.shape[1] > 1:
      # Multimetric (the DEFAULT algorithm must serve MO studies,
      # gp_ucb_pe.py:66,300): per-metric GPs + hypervolume-scalarized
      # UCB; the PE phase explores summed posterior stddev.
      warped = np.zeros_like(y_np)
      for m in range(y_np.shape[1]):
        warped[:, m] = output_warpers.create_default_warper().warp(
            y_np[:, m:m + 1]).flatten()
      prev = self._mo_posteriors or []
      if cfg.multitask_type != 'independent':
        from vizier_amd._src.algorithms.designers.gp_bandit import (
            _MultitaskTaskView,
        )
        from vizier_amd._src.gp import multitask
        kind = (multitask.MultiTaskType.SEPARABLE_DIAG
                if cfg.multitask_type == 'separable_diag'
                else multitask.MultiTaskType.SEPARABLE)
        y_all = torch.as_tensor(warped, dtype=cfg.dtype,
                                device=self._device)
        warm = self._mt_raw
        mt = multitask.train_multitask_gp(
            x, y_all, multitask_type=kind,
            num_restarts=cfg.ard_warm_restarts if warm is not None
            else cfg.ard_restarts,
            max_iters=cfg.ard_warm_iters if warm is not None
            else cfg.ard_max_iters,
            seed=self._seed, warm_start_raw=warm)
        self._mt_raw = mt.raw
        posts = [_MultitaskTaskView(mt, t)
                 for t in range(y_np.shape[1])]
      else:
        posts = []
        for m in range(y_np.shape[1]):
          y_m = torch.as_tensor(warped[:, m], dtype=cfg.dtype,
                                device=self._device)
          warm = prev[m].raw if m < len(prev) else None
          posts.append(gp_model.train_gp(
              x, y_m,
              num_restarts=cfg.ard_warm_restarts if warm is not None
              else cfg.ard_restarts,
              max_iters=cfg.ard_warm_iters if warm is not None
              else cfg.ard_max_iters,
              seed=self._seed + m, warm_start_raw=warm,
              ensemble_size=cfg.ensemble_size))
      self._mo_posteriors = posts
      self._posterior = posts[0]
      labels = torch.as_tensor(warped, dtype=cfg.dtype,
                               device=self._device)
      self._warped_labels = labels
      self._mo_scalarizer = acq_lib.create_hv_scalarization(
          cfg.num_scalarizations, y_np.shape[1], seed=self._seed,
          reference_point=acq_lib.get_reference_point(labels))
      self._last_fit_count = len(self._completed)
      return
    self._mo_posteriors = None

    warper = output_warpers.create_default_warper()
    y_np = warper.warp(y_np[:, :1]).flatten()
    y = torch.as_tensor(y_np, dtype=cfg.dtype, device=self._device)
    warm = self._posterior.raw if self._posterior is not None else None
    restarts = cfg.ard_warm_restarts if warm is not None \
        else cfg.ard_restarts
    iters = cfg.ard_warm_iters if warm is not None \
        else cfg.ard_max_iters
    if cfg.mixes_linear_kernel:
      from vizier_amd._src.gp import linear_matern
      self._posterior = linear_matern.train_linear_matern_gp(
          x, y, linear_coef=1.0, num_restarts=restarts,
          max_iters=iters, seed=self._seed, warm_start_raw=warm)
    else:
      self._posterior = gp_model.train_gp(
          x, y, num_restarts=restarts, max_iters=iters,
          seed=self._seed, warm_start_raw=warm,
          ensemble_size=cfg.ensemble_size)
    self._warped_labels = y
    self._last_fit_count = len(self._completed)

  def _choose_ucb(self, fresh_first: bool) -> bool:
    cfg = self._config
    params = self._posterior.params
    signal = float(params.amplitude) ** 2
    noise = float(params.noise)
    high_noise = (cfg.signal_to_noise_threshold > 0 and
                  signal / max(noise, 1e-12)
                  < cfg.signal_to_noise_threshold)
    if fresh_first:
      # New completed trials: exploit, unless PE-overwrite fires.
      p_pe = (cfg.pe_overwrite_probability_in_high_noise if high_noise
              else cfg.pe_overwrite_probability)
      return self._rng.random() >= p_pe
    # No new data (or later batch members): explore, unless UCB-overwrite.
    if high_noise:
      return False
    return self._rng.random() < cfg.ucb_overwrite_probability

  def _variance_posterior(self, x_all: torch.Tensor, posterior=None
                          ) -> gp_model.GPPosterior:
    """GP conditioned on all (completed+hallucinated) features.

    Only the predictive variance is used, so alpha is zeros.
    """
    posterior = posterior if posterior is not None else self._posterior
    params = posterior.params
    from vizier_amd._src.gp.matern import gram_matern52
    n = x_all.shape[0]
    from vizier_amd._src.gp import linear_matern
    if isinstance(posterior, linear_matern.LinearMaternPosterior):
      lc = posterior.linear_coef
      K = linear_matern._combined_gram(params, lc, x_all, None)
      K = K + params.noise * torch.eye(n, dtype=x_all.dtype,
                                       device=x_all.device)
      L = gp_model.cholesky_with_jitter(K, params.amplitude ** 2)
      return linear_matern.LinearMaternPosterior(
          x=x_all, params=params, linear_coef=lc, L=L,
          alpha=torch.zeros(n, dtype=x_all.dtype, device=x_all.device),
          nll=0.0)
    K = gram_matern52(x_all, None, params.lengthscales, params.amplitude)
    K = K + params.noise * torch.eye(n, dtype=x_all.dtype,
                                     device=x_all.device)
    L = gp_model.cholesky_with_jitter(K, params.amplitude ** 2)
    eye = torch.eye(n, dtype=x_all.dtype, device=x_all.device)
    z = torch.linalg.solve_triangular(L, eye, upper=False)
    K_inv = z.T @ z
    return gp_model.GPPosterior(
        x=x_all, params=params, L=L,
        alpha=torch.zeros(n, dtype=x_all.dtype, device=x_all.device),
        K_inv=K_inv, nll=0.0)

  def _optimize_set(self, x_all: torch.Tensor, q: int) -> torch.Tensor:
    """Jointly optimizes q exploration points via the SET-PE acquisition.

    Reference parity: SetPEScoreFunction (gp_ucb_pe.py:510-595).
    score(set) = logdet(cov(set | completed + pending))
                 + penalty_coef * sum_q min(explore_ucb(x_q) - thresh, 0)
    with the trust-region set penalty of gp_ucb_pe.py:245-269. Returns
    (q, D) feature rows.
    """
    cfg = self._config
    posterior = self._posterior
    if isinstance(posterior, gp_model.EnsembleGPPosterior):
      posterior = posterior.members[0]
    trust_region = acq_lib.TrustRegion.for_converter(
        x_all, self._converter) if cfg.use_trust_region else None

    with torch.no_grad():
      mean_obs, stddev_obs = posterior.predict(posterior.x)
      ucb_obs = mean_obs + cfg.ucb_coefficient * stddev_obs
      threshold = mean_obs[int(torch.argmax(ucb_obs))]
    var_post = self._variance_posterior(x_all, posterior)

    def score_fn(batch: CandidateBatch) -> torch.Tensor:
      xs = self._codec.decode(batch)                 # (B, q, D)
      B = xs.shape[0]
      flat = xs.reshape(B * q, -1)
      mean, stddev = posterior.predict(flat)
      explore_ucb = (mean + cfg.explore_region_ucb_coefficient *
                     stddev).reshape(B, q)
      penalty = cfg.cb_violation_penalty_coefficient * torch.minimum(
          explore_ucb - threshold,
          torch.zeros_like(explore_ucb)).sum(-1)
      _, cov = gp_bandit.posterior_batched_cov(var_post, xs)
      # Jitter scaled to the prior variance: near-duplicate rows (the
      # optimizer will try them) make cov singular; the reference maps
      # those to -inf via NaN logdet, cholesky_ex info does the same.
      amp2 = float(var_post.params.amplitude) ** 2
      eye = torch.eye(q, dtype=cov.dtype, device=cov.device)
      L, info = torch.linalg.cholesky_ex(cov + 1e-10 * amp2 * eye)
      logdet = 2.0 * torch.log(
          torch.diagonal(L, dim1=-2, dim2=-1).clamp_min(1e-30)).sum(-1)
      logdet = torch.where(info == 0, logdet,
                           torch.full_like(logdet, -float('inf')))
      scores = logdet + penalty
      if trust_region is not None:
        dist = trust_region.min_linf_distance(xs)    # (B, q)
        r = trust_region.trust_radius
        if r <= 0.5:
          scores = scores + torch.where(
              dist > r, -1e4 - dist, torch.zeros_like(dist)).sum(-1)
      return scores
    score_fn.graph_safe = False  # rocSOLVER cholesky: no capture

    factory = VectorizedOptimizerFactory(
        eagle_config=EagleStrategyConfig(),
        max_evaluations=cfg.max_evaluations,
        suggestion_batch_size=cfg.suggestion_batch_size)
    optimizer = factory(
        n_continuous=self._codec.n_continuous,
        categorical_sizes=self._codec.categorical_sizes, n_parallel=q,
        seed=self._seed + len(self._completed) + x_all.shape[0],
        device=self._device, dtype=cfg.dtype)
    results = optimizer.optimize(score_fn, count=1)
    return self._codec.decode(results.features)[0]   # (q, D)

  @staticmethod
  def _fusable(post, x_all: Optional[torch.Tensor] = None) -> bool:
    """Plain fp32 GPPosterior with K_inv on GPU (the HIP fused-scorer
    contract); when x_all is given, the posterior must be anchored on
    it so the kernel's trust-region distance is the right one."""
    return (isinstance(post, gp_model.GPPosterior) and
            post.K_inv is not None and post.x.is_cuda and
            post.x.dtype == torch.float32 and
            (x_all is None or post.x.data_ptr() == x_all.data_ptr()))

  def _tr_kernel_args(self, trust_region, device):
    onehot = (trust_region._onehot.to(torch.uint8)
              if trust_region is not None else torch.zeros(
                  self._converter.n_features, dtype=torch.uint8,
                  device=device))
    radius = (float(trust_region.trust_radius)
              if trust_region is not None else 0.0)
    return onehot, radius

  def _optimize_one(self, use_ucb: bool, x_all: torch.Tensor
                    ) -> torch.Tensor:
    cfg = self._config
    posterior = self._posterior
    trust_region = acq_lib.TrustRegion.for_converter(
        x_all, self._converter) if cfg.use_trust_region else None

    if self._mo_posteriors is not None:
      posts = self._mo_posteriors
      scalarizer = self._mo_scalarizer

      def _scalarized(xs: torch.Tensor, coef: float) -> torch.Tensor:
        per = [p_.predict(xs) for p_ in posts]
        ys = torch.stack([m_ + coef * s_ for m_, s_ in per], dim=-1)
        return scalarizer(ys).mean(dim=0)

      fused_mo = (cfg.dtype == torch.float32 and
                  all(self._fusable(p) for p in posts))
      if fused_mo:
        from vizier_amd._src.ops import dispatch as ops
        ext = ops.require_ext()
        weights = scalarizer.weights.to(x_all.device,
                                        torch.float32).contiguous()
        ref = scalarizer.reference_point
        ref = (ref.to(x_all.device, torch.float32).contiguous()
               if ref is not None else None)
        onehot, tr_radius = self._tr_kernel_args(trust_region,
                                                 x_all.device)

        # Scalars hoisted OUT of the closures: float(tensor) syncs the
        # device, which aborts hipGraph capture (measured: the stream-
        # capture failures on ROCm 7.2 were exactly these).
        post_amps = [float(p.params.amplitude) for p in posts]
        post_means = [float(p.params.mean) for p in posts]

        def _fused_scalarized(xs, coef, dist_src=None):
          means, sds, dist = [], [], None
          for i, p in enumerate(posts):
            m_, s_, d_ = ext.posterior_mean_std(
                xs, p.x, p.params.lengthscales, post_amps[i],
                post_means[i], p.alpha, p.K_inv, onehot)
            means.append(m_)
            sds.append(s_)
            dist = d_
          if dist_src is not None:
            dist = dist_src
          return means, sds, dist

      if use_ucb:
        if fused_mo and self._fusable(posts[0], x_all):
          # Fused MO UCB (same shape as GP-Bandit's config-5 path):
          # per-metric (mean, sd, dist) kernels + ONE scalarize+TR
          # launch; the posteriors anchor on x_all (no pending), so
          # the kernel trust-region distance is exact.
          def score_fn(batch: CandidateBatch) -> torch.Tensor:
            xs = self._codec.decode(batch)[:, 0, :]
            means, sds, dist = _fused_scalarized(
                xs, cfg.ucb_coefficient)
            return ext.hv_scalarize_tr(
                torch.stack(means), torch.stack(sds), weights, ref,
                dist if trust_region is not None else None,
                cfg.ucb_coefficient, tr_radius)
          score_fn.graph_safe = True
        else:
          def score_fn(batch: CandidateBatch) -> torch.Tensor:
            xs = self._codec.decode(batch)[:, 0, :]
            scores = _scalarized(xs, cfg.ucb_coefficient)
            if trust_region is not None:
              scores = trust_region.apply(xs, scores)
            return scores
          score_fn.graph_safe = False  # float(tensor) syncs in predict
      else:
        # Promising region: scalarized mean at the observed point with
        # the best scalarized UCB (multimetric analogue of
        # gp_ucb_pe.py:384 PEScoreFunction).
        with torch.no_grad():
          obs_ucb = _scalarized(posterior.x, cfg.ucb_coefficient)
          obs_mean = _scalarized(posterior.x, 0.0)
          threshold = obs_mean[int(torch.argmax(obs_ucb))]
        var_posts = [self._variance_posterior(x_all, p_) for p_ in posts]

        if fused_mo and all(self._fusable(vp, x_all)
                            for vp in var_posts):
          thr = float(threshold)
          pen_coef = cfg.cb_violation_penalty_coefficient
          vp_amps = [float(vp.params.amplitude) for vp in var_posts]

          def score_fn(batch: CandidateBatch) -> torch.Tensor:
            xs = self._codec.decode(batch)[:, 0, :]
            means, sds, _ = _fused_scalarized(
                xs, cfg.explore_region_ucb_coefficient)
            explore = ext.hv_scalarize_tr(
                torch.stack(means), torch.stack(sds), weights, ref,
                None, cfg.explore_region_ucb_coefficient, 0.0)
            stddev_sum, dist = None, None
            for vi, vp in enumerate(var_posts):
              _, s_all, d_ = ext.posterior_mean_std(
                  xs, vp.x, vp.params.lengthscales, vp_amps[vi], 0.0,
                  vp.alpha, vp.K_inv, onehot)
              stddev_sum = s_all if stddev_sum is None \
                  else stddev_sum + s_all
              dist = d_
            penalty = pen_coef * torch.minimum(
                explore - thr, torch.zeros_like(explore))
            scores = stddev_sum + penalty
            if trust_region is not None and tr_radius <= 0.5:
              scores = torch.where(dist <= tr_radius, scores,
                                   -1e4 - dist)
            return scores
          score_fn.graph_safe = True
        else:
          def score_fn(batch: CandidateBatch) -> torch.Tensor:
            xs = self._codec.decode(batch)[:, 0, :]
            explore = _scalarized(
                xs, cfg.explore_region_ucb_coefficient)
            stddev_sum = sum(vp.predict(xs)[1] for vp in var_posts)
            penalty = cfg.cb_violation_penalty_coefficient * \
                torch.minimum(explore - threshold,
                              torch.zeros_like(explore))
            scores = stddev_sum + penalty
            if trust_region is not None:
              scores = trust_region.apply(xs, scores)
            return scores
          score_fn.graph_safe = False  # float(tensor) syncs in predict
    elif use_ucb:
      scoring = acq_lib.ScoringFunction(
          posterior, acq_lib.UCB(cfg.ucb_coefficient), trust_region)

      def score_fn(batch: CandidateBatch) -> torch.Tensor:
        return scoring(self._codec.decode(batch)[:, 0, :])
      single_gp = isinstance(posterior, gp_model.GPPosterior)
      if single_gp and scoring._tr_anchored and \
          scoring._acq_name is not None:
        # No pending/hallucinated points: the trust region anchors at
        # the GP train set, so the pure-HIP chunked scorer applies —
        # hipGraph-capturable and megakernel-eligible (the common
        # sequential-suggest case; same fast path as GP-Bandit).
        score_fn.scoring = scoring
        score_fn.codec_identity = self._codec.identity
      else:
        # Pending points anchor the trust region elsewhere: composed
        # rocBLAS path, not capturable on this ROCm build.
        score_fn.graph_safe = False  # float(tensor) syncs in predict
    else:
      # Promising-region threshold: predicted mean at the observed point
      # with the highest UCB (gp_ucb_pe.py:175-205).
      mean_obs, stddev_obs = posterior.predict(posterior.x)
      ucb_obs = mean_obs + cfg.ucb_coefficient * stddev_obs
      threshold = mean_obs[int(torch.argmax(ucb_obs))]
      var_post = self._variance_posterior(x_all)

      if (cfg.dtype == torch.float32 and self._fusable(posterior) and
          self._fusable(var_post, x_all)):
        # Fused PE: two (mean, sd, dist) kernel calls + an elementwise
        # tail — hipGraph-capturable (the DEFAULT algorithm's batch
        # exploration phase was previously the ~20-launch eager chain).
        # The trust-region distance comes from the VAR posterior's
        # call: it anchors on x_all, exactly the region's trusted set.
        from vizier_amd._src.ops import dispatch as ops
        ext = ops.require_ext()
        onehot, tr_radius = self._tr_kernel_args(trust_region,
                                                 x_all.device)
        thr = float(threshold)
        pen_coef = cfg.cb_violation_penalty_coefficient
        exp_coef = cfg.explore_region_ucb_coefficient
        post = posterior
        # Hoisted: float(tensor) inside the closure syncs and aborts
        # stream capture.
        post_amp = float(post.params.amplitude)
        post_mean = float(post.params.mean)
        vp_amp = float(var_post.params.amplitude)

        def score_fn(batch: CandidateBatch) -> torch.Tensor:
          xs = self._codec.decode(batch)[:, 0, :]
          mean, stddev, _ = ext.posterior_mean_std(
              xs, post.x, post.params.lengthscales, post_amp,
              post_mean, post.alpha, post.K_inv, onehot)
          _, stddev_all, dist = ext.posterior_mean_std(
              xs, var_post.x, var_post.params.lengthscales, vp_amp,
              0.0, var_post.alpha, var_post.K_inv, onehot)
          explore_ucb = mean + exp_coef * stddev
          penalty = pen_coef * torch.minimum(
              explore_ucb - thr, torch.zeros_like(explore_ucb))
          scores = stddev_all + penalty
          if trust_region is not None and tr_radius <= 0.5:
            scores = torch.where(dist <= tr_radius, scores,
                                 -1e4 - dist)
          return scores
        score_fn.graph_safe = True
      else:
        def score_fn(batch: CandidateBatch) -> torch.Tensor:
          xs = self._codec.decode(batch)[:, 0, :]
          mean, stddev = posterior.predict(xs)
          explore_ucb = mean + \
              cfg.explore_region_ucb_coefficient * stddev
          _, stddev_all = var_post.predict(xs)
          penalty = cfg.cb_violation_penalty_coefficient * \
              torch.minimum(explore_ucb - threshold,
                            torch.zeros_like(explore_ucb))
          scores = stddev_all + penalty
          if trust_region is not None:
            scores = trust_region.apply(xs, scores)
          return scores
        score_fn.graph_safe = False  # float(tensor) syncs in predict

    factory = VectorizedOptimizerFactory(
        eagle_config=EagleStrategyConfig(),
        max_evaluations=cfg.max_evaluations,
        suggestion_batch_size=cfg.suggestion_batch_size)
    optimizer = factory(
        n_continuous=self._codec.n_continuous,
        categorical_sizes=self._codec.categorical_sizes,
        seed=self._seed + len(self._completed) + x_all.shape[0],
        device=self._device, dtype=cfg.dtype)

    if self._mo_posteriors is not None:
      # Eagle prior seeding needs scalar rewards: use the scalarized
      # warped labels (same scalarizer as the UCB phase).
      with torch.no_grad():
        rewards = self._mo_scalarizer(
            self._warped_labels).mean(dim=0).cpu().numpy()
    else:
      rewards = self._warped_labels.cpu().numpy()
    prior_features, prior_rewards = trials_to_sorted_features(
        self._converter, self._codec, self._completed, rewards,
        device=self._device, dtype=cfg.dtype)
    results = optimizer.optimize(score_fn, count=1,
                                 prior_features=prior_features,
                                 prior_rewards=prior_rewards)
    return self._codec.decode(results.features)[0, 0, :]
