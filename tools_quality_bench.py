"""Quality evidence for the batch (qEI) and multi-objective milestones.

Two experiments, both on shifted BBOB at D=20, GPU GP-Bandit vs
quasi-random at the SAME trial budget:

1. qEI batches: suggest(q=4) x 25 rounds (100 trials). Checks that the
   q-EI fantasy batching produces DISTINCT, useful points (config-3
   milestone) — best-value-found comparison.
2. Multi-objective: two conflicting sphere objectives, 100 trials;
   GP-Bandit's hypervolume-scalarized path (config-5 milestone) vs
   quasi-random, scored by dominated hypervolume of the final Pareto
   front (ParetoFrontier randomized estimator, reference
   multimetric semantics).

Usage: python tools_quality_bench.py [--out profiles/quality.json]
"""

import argparse
import json
import sys
import time

import numpy as np

sys.path.insert(0, '.')

import torch

from vizier_amd import pyvizier as vz
from vizier_amd._src.algorithms.core.abstractions import (
    ActiveTrials,
    CompletedTrials,
)
from vizier_amd._src.benchmarks.experimenters.synthetic import bbob
from vizier_amd._src.pyvizier import multimetric


def _problem(dim, metric_names):
  problem = vz.ProblemStatement()
  root = problem.search_space.root
  for i in range(dim):
    root.add_float_param(f'x{i}', -5.0, 5.0)
  problem.metric_information = vz.MetricsConfig([
      vz.MetricInformation(name=n, goal=vz.ObjectiveMetricGoal.MAXIMIZE)
      for n in metric_names])
  return problem


def _gp_designer(problem, seed, evals, batch):
  from vizier_amd._src.algorithms.designers.gp_bandit import (
      GPBanditConfig, VizierGPBandit)
  acq = 'qei' if batch > 1 else 'ucb'
  return VizierGPBandit(problem, GPBanditConfig(
      max_evaluations=evals, acquisition=acq,
      ard_restarts=4, ard_max_iters=30, device='cuda'), seed=seed)


def _qr_designer(problem, seed):
  from vizier_amd._src.algorithms.designers.quasi_random import (
      QuasiRandomDesigner)
  return QuasiRandomDesigner(problem.search_space, seed=seed)


def run_qei(fn, dim, rounds, q, seed, use_gp, evals):
  problem = _problem(dim, ['obj'])
  designer = (_gp_designer(problem, seed, evals, q) if use_gp
              else _qr_designer(problem, seed))
  shift = np.random.default_rng(2000 + seed).uniform(-2.0, 2.0, dim)
  best, uid = np.inf, 0
  for _ in range(rounds):
    trials = []
    for s in designer.suggest(q):
      uid += 1
      x = np.array([s.parameters.get_value(f'x{i}') for i in range(dim)])
      v = fn(x - shift, seed=seed)
      best = min(best, v)
      t = s.to_trial(uid)
      t.complete(vz.Measurement(metrics={'obj': -v}))
      trials.append(t)
    designer.update(CompletedTrials(trials), ActiveTrials())
  return float(best)


def run_mo(dim, n_trials, seed, use_gp, evals):
  problem = _problem(dim, ['f1', 'f2'])
  designer = (_gp_designer(problem, seed, evals, 1) if use_gp
              else _qr_designer(problem, seed))
  rng = np.random.default_rng(3000 + seed)
  c1 = rng.uniform(-1.5, 1.5, dim)
  c2 = rng.uniform(-1.5, 1.5, dim)
  ys, uid = [], 0
  for _ in range(n_trials):
    for s in designer.suggest(1):
      uid += 1
      x = np.array([s.parameters.get_value(f'x{i}') for i in range(dim)])
      # Conflicting objectives (maximize): peak at c1 resp. c2.
      f1 = -float(((x - c1) ** 2).sum()) / dim
      f2 = -float(((x - c2) ** 2).sum()) / dim
      ys.append([f1, f2])
      t = s.to_trial(uid)
      t.complete(vz.Measurement(metrics={'f1': f1, 'f2': f2}))
      designer.update(CompletedTrials([t]), ActiveTrials())
  pts = np.asarray(ys)
  origin = np.array([-30.0, -30.0])  # dominated reference point
  front = multimetric.ParetoFrontier(pts, origin, num_vectors=10000)
  return float(front.hypervolume())


def main():
  parser = argparse.ArgumentParser()
  parser.add_argument('--dim', type=int, default=20)
  parser.add_argument('--evals', type=int, default=10000)
  parser.add_argument('--seeds', type=int, default=2)
  parser.add_argument('--out', default=None)
  parser.add_argument('--only', choices=['qei', 'mo'], default=None)
  args = parser.parse_args()

  results = {'qei': {}, 'mo': {}}
  fns = {'sphere': bbob.Sphere, 'rastrigin': bbob.Rastrigin}
  if args.only == 'mo':
    fns = {}
  for name, fn in fns.items():
    for algo, use_gp in (('gp_qei_q4', True), ('quasi_random', False)):
      vals = []
      for seed in range(args.seeds):
        t0 = time.time()
        v = run_qei(fn, args.dim, rounds=25, q=4, seed=seed,
                    use_gp=use_gp, evals=args.evals)
        vals.append(v)
        print(f'qei {name} {algo} seed={seed}: best={v:.4f} '
              f'({time.time() - t0:.1f}s)', flush=True)
      results['qei'][f'{name}/{algo}'] = vals

  mo_algos = [] if args.only == 'qei' else [
      ('gp_hv_scalarized', True), ('quasi_random', False)]
  for algo, use_gp in mo_algos:
    vals = []
    for seed in range(args.seeds):
      t0 = time.time()
      hv = run_mo(args.dim, n_trials=100, seed=seed, use_gp=use_gp,
                  evals=args.evals)
      vals.append(hv)
      print(f'mo {algo} seed={seed}: hypervolume={hv:.4f} '
            f'({time.time() - t0:.1f}s)', flush=True)
    results['mo'][algo] = vals

  print(json.dumps(results))
  if args.out:
    with open(args.out, 'w') as f:
      json.dump(results, f, indent=2)


if __name__ == '__main__':
  main()
