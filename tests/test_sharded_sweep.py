"""Multi-process tests for the RCCL/xGMI sharded sweep (gloo on CPU)."""

import os

import numpy as np
import pytest
import torch
import torch.multiprocessing as mp


def _worker_topk(rank, world_size, port, results):
  import torch.distributed as dist
  from vizier_amd._src.parallel import sharded_sweep
  dist.init_process_group(
      backend='gloo', init_method=f'tcp://127.0.0.1:{port}',
      rank=rank, world_size=world_size)
  try:
    # Each rank has its own candidates; global top-2 must be identical
    # on every rank.
    features = torch.arange(3, dtype=torch.float32).reshape(3, 1) + \
        10.0 * rank
    rewards = features[:, 0].clone()
    top_f, top_r = sharded_sweep.allgather_topk(features, rewards, 2)
    results[rank] = (top_f.numpy().tolist(), top_r.numpy().tolist())
  finally:
    dist.destroy_process_group()


def _worker_designer(rank, world_size, port, results):
  import torch.distributed as dist
  from vizier_amd import pyvizier as vz
  from vizier_amd._src.algorithms.core.abstractions import (
      ActiveTrials,
      CompletedTrials,
  )
  from vizier_amd._src.algorithms.designers.gp_bandit import (
      GPBanditConfig,
      VizierGPBandit,
  )
  dist.init_process_group(
      backend='gloo', init_method=f'tcp://127.0.0.1:{port}',
      rank=rank, world_size=world_size)
  try:
    problem = vz.ProblemStatement()
    for i in range(3):
      problem.search_space.root.add_float_param(f'x{i}', 0.0, 1.0)
    problem.metric_information.append(vz.MetricInformation(
        name='obj', goal=vz.ObjectiveMetricGoal.MAXIMIZE))
    designer = VizierGPBandit(problem, GPBanditConfig(
        max_evaluations=300, ard_restarts=1, ard_max_iters=10,
        device='cpu', data_parallel=True), seed=0)
    rng = np.random.default_rng(0)  # identical data on all ranks
    trials = []
    for uid in range(1, 9):
      params = {f'x{i}': float(v)
                for i, v in enumerate(rng.uniform(0, 1, 3))}
      t = vz.Trial(params, id=uid)
      x = np.array(list(params.values()))
      t.complete(vz.Measurement(metrics={'obj': float(-(x ** 2).sum())}))
      trials.append(t)
    designer.update(CompletedTrials(trials), ActiveTrials())
    suggestion = designer.suggest(1)[0]
    results[rank] = dict(suggestion.parameters.as_dict())
  finally:
    dist.destroy_process_group()


def _worker_topk_ties(rank, world_size, port, results):
  import torch.distributed as dist
  from vizier_amd._src.parallel import sharded_sweep
  dist.init_process_group(
      backend='gloo', init_method=f'tcp://127.0.0.1:{port}',
      rank=rank, world_size=world_size)
  try:
    # EVERY rank reports the same reward values but different features:
    # the global top-k must still be identical on all ranks (stable
    # sort over the gathered order breaks ties by rank then index).
    rewards = torch.tensor([5.0, 5.0, 3.0])
    features = torch.full((3, 2), float(rank))
    top_f, top_r = sharded_sweep.allgather_topk(features, rewards, 4)
    results[rank] = (top_f.numpy().tolist(), top_r.numpy().tolist())
  finally:
    dist.destroy_process_group()


def _worker_linear_broadcast(rank, world_size, port, results):
  import torch.distributed as dist
  from vizier_amd._src.gp import linear_matern
  from vizier_amd._src.parallel import sharded_sweep
  dist.init_process_group(
      backend='gloo', init_method=f'tcp://127.0.0.1:{port}',
      rank=rank, world_size=world_size)
  try:
    # Rank 0 fits the linear+Matern GP; other ranks fit on SHUFFLED
    # labels (different posterior). After broadcast_posterior all ranks
    # must predict identically, including the linear slope/shift aux.
    g = torch.Generator().manual_seed(7)
    x = torch.rand(12, 3, generator=g)
    y = (2.0 * x[:, 0] - x[:, 1]).contiguous()
    if rank != 0:
      y = y.flip(0).contiguous()
    post = linear_matern.train_linear_matern_gp(
        x, y, linear_coef=1.0, num_restarts=1, max_iters=10, seed=3)
    sharded_sweep.broadcast_posterior(post)
    xq = torch.linspace(0, 1, 5).reshape(5, 1).repeat(1, 3)
    mean, std = post.predict(xq)
    results[rank] = (mean.detach().numpy().tolist(),
                     std.detach().numpy().tolist(),
                     float(post.params.slope), float(post.params.shift))
  finally:
    dist.destroy_process_group()


def _run_multiproc(target, world_size=2):
  ctx = mp.get_context('spawn')
  manager = ctx.Manager()
  results = manager.dict()
  port = 29600 + os.getpid() % 1000
  procs = [ctx.Process(target=target,
                       args=(r, world_size, port, results))
           for r in range(world_size)]
  for p in procs:
    p.start()
  for p in procs:
    p.join(timeout=180)
    assert p.exitcode == 0, f'worker failed with {p.exitcode}'
  return dict(results)


class TestShardedSweep:

  def test_allgather_topk_deterministic_across_ranks(self):
    results = _run_multiproc(_worker_topk)
    assert results[0] == results[1]
    # Global best comes from rank 1 (values 10, 11, 12).
    assert results[0][1] == [12.0, 11.0]

  def test_data_parallel_designer_identical_suggestions(self):
    results = _run_multiproc(_worker_designer)
    assert results[0] == results[1]

  def test_allgather_topk_ws8(self):
    # The driver's scaling bench runs DP=8: same all-gather shape here.
    results = _run_multiproc(_worker_topk, world_size=8)
    for r in range(1, 8):
      assert results[r] == results[0]
    # Global best values come from rank 7 (70, 71, 72).
    assert results[0][1] == [72.0, 71.0]

  def test_allgather_topk_reward_ties_deterministic(self):
    # Equal rewards on every rank: selection must not depend on which
    # rank evaluates it (stable order = rank-major, index-minor).
    results = _run_multiproc(_worker_topk_ties, world_size=4)
    for r in range(1, 4):
      assert results[r] == results[0]
    # Ties broken by gather order: rank 0's two 5.0s, then rank 1's.
    assert results[0][1] == [5.0, 5.0, 5.0, 5.0]
    assert results[0][0] == [[0.0, 0.0], [0.0, 0.0],
                             [1.0, 1.0], [1.0, 1.0]]

  def test_linear_kernel_posterior_broadcast(self):
    results = _run_multiproc(_worker_linear_broadcast)
    m0, s0, slope0, shift0 = results[0]
    m1, s1, slope1, shift1 = results[1]
    np.testing.assert_allclose(m0, m1, rtol=1e-6)
    np.testing.assert_allclose(s0, s1, rtol=1e-6)
    assert slope0 == pytest.approx(slope1)
    assert shift0 == pytest.approx(shift1)
