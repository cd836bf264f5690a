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
