"""Simple regret @ 100 trials on BBOB: GP designers vs quasi-random.

The BASELINE.json north-star quality metric. Runs each (function,
algorithm) pair for `--trials` sequential trials and reports the best
value found (functions are minimized; regret = best - f_opt, and for
shifted/rotated BBOB functions here f_opt = f(x_opt) = 0 at the origin
for Sphere/Rastrigin-style members used below).

Usage: python tools_regret_bench.py [--trials 100] [--dim 20]
       [--evals 10000] [--out profiles/regret.json]
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


def run(designer_factory, fn, dim, n_trials, seed):
  problem = bbob.DefaultBBOBProblemStatement(dim)
  # Designers maximize; flip the sign.
  problem.metric_information = vz.MetricsConfig([vz.MetricInformation(
      name='bbob_eval', goal=vz.ObjectiveMetricGoal.MAXIMIZE)])
  designer = designer_factory(problem, seed)
  # Shift the optimum away from the domain center (standard BBOB
  # practice; the center-default seed would otherwise hit x_opt = 0).
  shift = np.random.default_rng(1000 + seed).uniform(-2.0, 2.0, dim)
  best = np.inf
  uid = 0
  for _ in range(n_trials):
    for s in designer.suggest(1):
      uid += 1
      x = np.array([s.parameters.get_value(f'x{i}') for i in range(dim)])
      value = fn(x - shift, seed=seed)
      best = min(best, value)
      t = s.to_trial(uid)
      t.complete(vz.Measurement(metrics={'bbob_eval': -value}))
      designer.update(CompletedTrials([t]), ActiveTrials())
  return float(best)


def main():
  parser = argparse.ArgumentParser()
  parser.add_argument('--trials', type=int, default=100)
  parser.add_argument('--dim', type=int, default=20)
  parser.add_argument('--evals', type=int, default=10000)
  parser.add_argument('--seeds', type=int, default=1)
  parser.add_argument('--out', default=None)
  args = parser.parse_args()

  device = 'cuda' if torch.cuda.is_available() else 'cpu'

  def gp_bandit(problem, seed):
    from vizier_amd._src.algorithms.designers.gp_bandit import (
        GPBanditConfig,
        VizierGPBandit,
    )
    return VizierGPBandit(problem, GPBanditConfig(
        max_evaluations=args.evals, ard_restarts=2, ard_max_iters=30,
        ard_warm_iters=15, device=device), seed=seed)

  def gp_ucb_pe(problem, seed):
    from vizier_amd._src.algorithms.designers.gp_ucb_pe import (
        UCBPEConfig,
        VizierGPUCBPEBandit,
    )
    return VizierGPUCBPEBandit(problem, UCBPEConfig(
        max_evaluations=args.evals, ard_restarts=2, ard_max_iters=30,
        ard_warm_iters=15, device=device), seed=seed)

  def quasi_random(problem, seed):
    from vizier_amd._src.algorithms.designers.quasi_random import (
        QuasiRandomDesigner,
    )
    return QuasiRandomDesigner(problem.search_space, seed=seed)

  def gp_bandit_fp64(problem, seed):
    from vizier_amd._src.algorithms.designers.gp_bandit import (
        GPBanditConfig,
        VizierGPBandit,
    )
    return VizierGPBandit(problem, GPBanditConfig(
        max_evaluations=args.evals, ard_restarts=2, ard_max_iters=30,
        ard_warm_iters=15, device=device,
        dtype=torch.float64), seed=seed)

  functions = {'Sphere': bbob.Sphere, 'Rastrigin': bbob.Rastrigin,
               'SharpRidge': bbob.SharpRidge,
               'Rosenbrock': bbob.Rosenbrock,
               'Ellipsoidal': bbob.Ellipsoidal,
               'AttractiveSector': bbob.AttractiveSector}
  algorithms = {'gp_bandit_ucb': gp_bandit, 'gp_ucb_pe': gp_ucb_pe,
                'gp_bandit_fp64': gp_bandit_fp64,
                'quasi_random': quasi_random}
  results = {}
  for fname, fn in functions.items():
    for aname, factory in algorithms.items():
      bests = []
      for seed in range(args.seeds):
        t0 = time.time()
        best = run(factory, fn, args.dim, args.trials, seed)
        bests.append(best)
        print(f'{fname:12s} {aname:14s} seed={seed} '
              f'best@{args.trials}={best:.4f} '
              f'({time.time()-t0:.1f}s)', flush=True)
      results[f'{fname}/{aname}'] = bests
  if args.out:
    with open(args.out, 'w') as f:
      json.dump({'dim': args.dim, 'trials': args.trials,
                 'acquisition_evals': args.evals,
                 'best_value_minimized': results}, f, indent=2)
    print('wrote', args.out)


if __name__ == '__main__':
  main()
