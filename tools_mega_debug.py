"""Megakernel debug: crash isolation + per-phase timing.

Runs each probe in a SUBPROCESS so an async HIP fault in one doesn't
poison the next. Usage: python tools_mega_debug.py <probe>
  a: train_gp N=300 WITHOUT importing the HIP extension
  b: import ext, then train_gp N=300
  c: direct eagle_sweep micro vs hipGraph-equivalent manual loop
  d: eagle_sweep iteration timing at bench shapes (N=1000, D=20, B=25)
"""

import subprocess
import sys
import time

sys.path.insert(0, '.')


def probe_a():
  import torch
  from vizier_amd._src.gp import gp_model
  g = torch.Generator().manual_seed(0)
  x = torch.rand(300, 12, generator=g).cuda()
  y = (-((x - 0.4) ** 2).sum(-1) + 0.01 * torch.randn(300, generator=g).cuda())
  post = gp_model.train_gp(x, y, num_restarts=2, max_iters=15)
  torch.cuda.synchronize()
  print('A OK: nll', post.nll, flush=True)


def probe_b():
  import torch
  from vizier_amd._src.ops import dispatch
  ext = dispatch.require_ext()
  from vizier_amd._src.gp import gp_model
  g = torch.Generator().manual_seed(0)
  x = torch.rand(300, 12, generator=g).cuda()
  y = (-((x - 0.4) ** 2).sum(-1) + 0.01 * torch.randn(300, generator=g).cuda())
  post = gp_model.train_gp(x, y, num_restarts=2, max_iters=15)
  torch.cuda.synchronize()
  print('B OK: nll', post.nll, flush=True)


def _setup_sweep():
  import torch
  from vizier_amd._src.gp import acquisitions as acq_lib
  from vizier_amd._src.gp import gp_model
  from vizier_amd._src.algorithms.optimizers.eagle import (
      EagleStrategyConfig)
  from vizier_amd._src.algorithms.optimizers.vectorized import (
      VectorizedOptimizerFactory)
  g = torch.Generator().manual_seed(0)
  x = torch.rand(300, 12, generator=g).cuda()
  y = (-((x - 0.4) ** 2).sum(-1) + 0.01 * torch.randn(300, generator=g).cuda())
  post = gp_model.train_gp(x, y, num_restarts=2, max_iters=15)
  onehot = torch.zeros(12, dtype=torch.bool, device='cuda')
  tr = acq_lib.TrustRegion(post.x, onehot)
  scoring = acq_lib.ScoringFunction(post, acq_lib.UCB(coefficient=1.8), tr)
  factory = VectorizedOptimizerFactory(
      eagle_config=EagleStrategyConfig(), max_evaluations=5000,
      suggestion_batch_size=25)

  def make_opt():
    return factory(n_continuous=12, categorical_sizes=[], n_parallel=1,
                   seed=7, device='cuda', dtype=torch.float32)

  def score_fn(batch):
    return scoring(batch.continuous[:, 0, :])
  score_fn.scoring = scoring
  score_fn.codec_identity = True
  return make_opt, score_fn


def probe_c():
  import torch
  make_opt, score_fn = _setup_sweep()
  opt_g = make_opt()
  opt_g._megakernel_applicable = lambda s: False
  res_g = opt_g.optimize(score_fn, count=3)
  torch.cuda.synchronize()
  print('graph path done, used_graph =', opt_g.last_used_graph,
        opt_g.last_graph_error, flush=True)
  opt_m = make_opt()
  res_m = opt_m.optimize(score_fn, count=3)
  torch.cuda.synchronize()
  print('mega path done, used_mega =', opt_m.last_used_megakernel,
        opt_m.last_graph_error, flush=True)
  print('rewards equal:', torch.equal(res_m.rewards, res_g.rewards))
  print('graph rewards:', res_g.rewards.tolist())
  print('mega  rewards:', res_m.rewards.tolist())
  d = (res_m.features.continuous - res_g.features.continuous).abs().max()
  print('feature max diff:', float(d))


def probe_d():
  import torch
  from vizier_amd._src.ops import dispatch
  ext = dispatch.require_ext()
  from vizier_amd._src.gp import acquisitions as acq_lib
  from vizier_amd._src.gp import gp_model
  from vizier_amd._src.algorithms.optimizers.eagle import (
      EagleStrategyConfig, VectorizedEagleStrategy)
  g = torch.Generator().manual_seed(0)
  x = torch.rand(1000, 20, generator=g).cuda()
  y = torch.randn(1000, generator=g).cuda()
  post = gp_model.train_gp(x, y, num_restarts=1, max_iters=5)
  strategy = VectorizedEagleStrategy(
      n_continuous=20, categorical_sizes=[], batch_size=25,
      config=EagleStrategyConfig(), n_parallel=1, seed=3, device='cuda')
  state = strategy.init_state()
  state.rewards.normal_()
  state.iterations = strategy.pool_size // 25
  strategy._iter_t.fill_(state.iterations)
  b, n = 25, 1000
  inv_ls = (1.0 / post.params.lengthscales).contiguous()
  k_ws = torch.empty(b, n, device='cuda')
  mu_ws = torch.empty(b, device='cuda')
  dist_ws = torch.empty(b, device='cuda')
  var_ws = torch.empty(b, 10, device='cuda')
  barrier_buf = torch.zeros(2, dtype=torch.int32, device='cuda')
  cfg = strategy.config

  def run(iters):
    return ext.eagle_sweep(
        state.continuous, state.rewards, state.perturbations,
        state.best_reward.reshape(1), strategy._iter_t, barrier_buf,
        post.x, inv_ls,
        post.alpha, post.K_inv, strategy._out_cont, k_ws, mu_ws,
        dist_ws, var_ws, strategy.pool_size // 25, b,
        strategy.pool_size, state.iterations, iters, cfg.visibility,
        cfg.gravity, cfg.negative_gravity, cfg.normalization_scale,
        cfg.penalize_factor, cfg.perturbation_lower_bound,
        cfg.perturbation, strategy._seed, strategy._seed ^ 0xABCDEF,
        1.0, 0.0, 0, 1.8, 0.0, 0.24)

  grid = run(10)
  torch.cuda.synchronize()
  print('grid =', grid, flush=True)
  for iters in (100, 1000):
    t0 = time.perf_counter()
    run(iters)
    torch.cuda.synchronize()
    dt = time.perf_counter() - t0
    print(f'{iters} iters: {dt*1e3:.1f} ms = {dt/iters*1e6:.1f} us/iter',
          flush=True)


def probe_e():
  """Bisect: first in-kernel iteration where megakernel diverges from
  the eager ext path (bitwise — eager steady-state uses the same
  standalone kernels the graph captures)."""
  import torch
  from vizier_amd._src.ops import dispatch
  ext = dispatch.require_ext()
  from vizier_amd._src.gp import acquisitions as acq_lib
  from vizier_amd._src.gp import gp_model
  from vizier_amd._src.algorithms.optimizers.eagle import (
      EagleStrategyConfig, VectorizedEagleStrategy)
  from vizier_amd._src.ops import dispatch as ops
  g = torch.Generator().manual_seed(0)
  x = torch.rand(300, 12, generator=g).cuda()
  y = (-((x - 0.4) ** 2).sum(-1) + 0.01 * torch.randn(300, generator=g).cuda())
  post = gp_model.train_gp(x, y, num_restarts=2, max_iters=15)
  onehot = torch.zeros(12, dtype=torch.bool, device='cuda')
  tr = acq_lib.TrustRegion(post.x, onehot)
  scoring = acq_lib.ScoringFunction(post, acq_lib.UCB(coefficient=1.8), tr)

  def score_fn(batch):
    return scoring(batch.continuous[:, 0, :])

  def make(seed=7):
    s = VectorizedEagleStrategy(
        n_continuous=12, categorical_sizes=[], batch_size=25,
        config=EagleStrategyConfig(), n_parallel=1, seed=seed,
        device='cuda')
    st = s.init_state()
    nb = s.pool_size // s.batch_size
    for _ in range(nb + 2):
      b = s.suggest(st)
      r = score_fn(b).detach()
      s.update(st, b, r)
    return s, st

  for T in (1, 2, 3, 5, 10, 30, 100):
    sA, stA = make()
    for _ in range(T):
      b = sA.suggest(stA)
      r = score_fn(b).detach()
      sA.update(stA, b, r)
    sB, stB = make()
    n = post.x.shape[0]
    bsz = sB.batch_size
    inv_ls = (1.0 / post.params.lengthscales).contiguous()
    k_ws = torch.empty(bsz, n, device='cuda')
    mu_ws = torch.empty(bsz, device='cuda')
    dist_ws = torch.empty(bsz, device='cuda')
    var_ws = torch.empty(bsz, 10, device='cuda')
    bar = torch.zeros(2, dtype=torch.int32, device='cuda')
    cfg = sB.config
    ext.eagle_sweep(
        stB.continuous, stB.rewards, stB.perturbations,
        stB.best_reward.reshape(1), sB._iter_t, bar, post.x, inv_ls,
        post.alpha, post.K_inv, sB._out_cont, k_ws, mu_ws, dist_ws,
        var_ws, sB.pool_size // bsz, bsz, sB.pool_size,
        stB.iterations, T, cfg.visibility, cfg.gravity,
        cfg.negative_gravity, cfg.normalization_scale,
        cfg.penalize_factor, cfg.perturbation_lower_bound,
        cfg.perturbation, sB._seed, sB._seed ^ 0xABCDEF,
        scoring._amp * scoring._amp, scoring._mean_c,
        ops.ACQ_CODES['ucb'], scoring._coef, scoring._best,
        scoring._tr_radius)
    torch.cuda.synchronize()
    pe = torch.equal(stA.continuous, stB.continuous)
    re = torch.equal(stA.rewards, stB.rewards)
    if pe and re:
      print(f'T={T}: identical', flush=True)
    else:
      dr = (stA.rewards - stB.rewards).abs()
      dp = (stA.continuous - stB.continuous).abs().amax(dim=(1, 2))
      bad = torch.nonzero(dr + dp > 0).flatten().tolist()
      print(f'T={T}: DIVERGED slots={bad[:8]} '
            f'dr={[round(float(dr[i]),6) for i in bad[:4]]} '
            f'dp={[round(float(dp[i]),6) for i in bad[:4]]} ', flush=True)
      print('  A rewards:', [round(float(stA.rewards[i]), 6) for i in bad[:4]])
      print('  B rewards:', [round(float(stB.rewards[i]), 6) for i in bad[:4]])
      break


def probe_f():
  """One in-kernel iteration; compare suggest output and scores against
  the eager ext path phase by phase."""
  import torch
  from vizier_amd._src.ops import dispatch
  ext = dispatch.require_ext()
  from vizier_amd._src.gp import acquisitions as acq_lib
  from vizier_amd._src.gp import gp_model
  from vizier_amd._src.algorithms.optimizers.eagle import (
      EagleStrategyConfig, VectorizedEagleStrategy)
  from vizier_amd._src.ops import dispatch as ops
  g = torch.Generator().manual_seed(0)
  x = torch.rand(300, 12, generator=g).cuda()
  y = (-((x - 0.4) ** 2).sum(-1) + 0.01 * torch.randn(300, generator=g).cuda())
  post = gp_model.train_gp(x, y, num_restarts=2, max_iters=15)
  onehot = torch.zeros(12, dtype=torch.bool, device='cuda')
  tr = acq_lib.TrustRegion(post.x, onehot)
  scoring = acq_lib.ScoringFunction(post, acq_lib.UCB(coefficient=1.8), tr)

  def score_fn(batch):
    return scoring(batch.continuous[:, 0, :])

  def make(seed=7):
    s = VectorizedEagleStrategy(
        n_continuous=12, categorical_sizes=[], batch_size=25,
        config=EagleStrategyConfig(), n_parallel=1, seed=seed,
        device='cuda')
    st = s.init_state()
    nb = s.pool_size // s.batch_size
    for _ in range(nb + 2):
      b = s.suggest(st)
      r = score_fn(b).detach()
      s.update(st, b, r)
    return s, st

  # Eager reference for ONE more iteration (keep copies BEFORE).
  sA, stA = make()
  batch = sA.suggest(stA)
  xq = batch.continuous[:, 0, :].clone()
  rewards_eager = score_fn(batch).detach().clone()
  sA.update(stA, batch, rewards_eager)

  # Megakernel ONE iteration from the same warm state.
  sB, stB = make()
  n, bsz = 300, 25
  inv_ls = (1.0 / post.params.lengthscales).contiguous()
  k_ws = torch.empty(bsz, n, device='cuda')
  mu_ws = torch.empty(bsz, device='cuda')
  dist_ws = torch.empty(bsz, device='cuda')
  var_ws = torch.empty(bsz, 10, device='cuda')
  bar = torch.zeros(2, dtype=torch.int32, device='cuda')
  cfg = sB.config
  ext.eagle_sweep(
      stB.continuous, stB.rewards, stB.perturbations,
      stB.best_reward.reshape(1), sB._iter_t, bar, post.x, inv_ls,
      post.alpha, post.K_inv, sB._out_cont, k_ws, mu_ws, dist_ws,
      var_ws, sB.pool_size // bsz, bsz, sB.pool_size,
      stB.iterations, 1, cfg.visibility, cfg.gravity,
      cfg.negative_gravity, cfg.normalization_scale,
      cfg.penalize_factor, cfg.perturbation_lower_bound,
      cfg.perturbation, sB._seed, sB._seed ^ 0xABCDEF,
      scoring._amp * scoring._amp, scoring._mean_c,
      ops.ACQ_CODES['ucb'], scoring._coef, scoring._best,
      scoring._tr_radius)
  torch.cuda.synchronize()
  out_mega = sB._out_cont[:, 0, :] if sB._out_cont.dim() == 3 else       sB._out_cont
  print('suggest equal:', torch.equal(out_mega, xq),
        'maxdiff', float((out_mega - xq).abs().max()))
  # Scores: recompute finalize from the megakernel partials.
  var = var_ws.sum(-1)
  amp2 = scoring._amp * scoring._amp
  sd = (amp2 - var).clamp_min(1e-12).sqrt()
  mu = mu_ws + scoring._mean_c
  sc = mu + 1.8 * sd
  pen = dist_ws > scoring._tr_radius
  sc = torch.where(pen, -1e4 - dist_ws, sc)
  print('scores equal:', torch.equal(sc, rewards_eager),
        'maxdiff', float((sc - rewards_eager).abs().max()))
  print('pool equal:', torch.equal(stA.continuous, stB.continuous),
        'rewards equal:', torch.equal(stA.rewards, stB.rewards),
        'rdiff', float((stA.rewards - stB.rewards).abs().max()))


if __name__ == '__main__':
  which = sys.argv[1] if len(sys.argv) > 1 else 'all'
  if which == 'all':
    for p in 'abcde':
      r = subprocess.run(
          ['timeout', '180', sys.executable, __file__, p],
          capture_output=True, text=True)
      print(f'--- probe {p} (rc={r.returncode}) ---')
      print(r.stdout[-2000:])
      if r.returncode != 0:
        print('STDERR:', r.stderr[-1500:])
  else:
    globals()[f'probe_{which}']()
