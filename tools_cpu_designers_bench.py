"""Breadth evidence: every non-GPU designer family on shifted BBOB.

best-value@100-trials, 8D Sphere + Rastrigin, seed 0 (CPU only).
"""
import json
import sys
import time

import numpy as np

sys.path.insert(0, '.')
from vizier_amd import pyvizier as vz
from vizier_amd._src.algorithms.core.abstractions import (
    ActiveTrials, CompletedTrials)
from vizier_amd._src.benchmarks.experimenters.synthetic import bbob

DIM = 8


def problem():
  p = vz.ProblemStatement()
  for i in range(DIM):
    p.search_space.root.add_float_param(f'x{i}', -5.0, 5.0)
  p.metric_information.append(vz.MetricInformation(
      name='obj', goal=vz.ObjectiveMetricGoal.MAXIMIZE))
  return p


def designers():
  from vizier_amd._src.algorithms.designers.random import RandomDesigner
  from vizier_amd._src.algorithms.designers.quasi_random import (
      QuasiRandomDesigner)
  from vizier_amd._src.algorithms.designers.cmaes import CMAESDesigner
  from vizier_amd._src.algorithms.designers.eagle_strategy import (
      eagle_strategy)
  from vizier_amd._src.algorithms.designers.gp_bandit import (
      GPBanditConfig, VizierGPBandit)

  return {
      'random': lambda p: RandomDesigner(p.search_space, seed=0),
      'quasi_random': lambda p: QuasiRandomDesigner(p.search_space, seed=0),
      'cmaes': lambda p: CMAESDesigner(p, seed=0),
      'eagle': lambda p: eagle_strategy.EagleStrategyDesigner(p, seed=0),
      'gp_bandit_cpu': lambda p: VizierGPBandit(p, GPBanditConfig(
          max_evaluations=2000, ard_restarts=2, ard_max_iters=20,
          device='cpu'), seed=0),
  }


def run(factory, fn, trials=100, seed=0):
  p = problem()
  d = factory(p)
  shift = np.random.default_rng(500 + seed).uniform(-2, 2, DIM)
  best, uid = np.inf, 0
  for _ in range(trials):
    for s in d.suggest(1):
      uid += 1
      x = np.array([s.parameters.get_value(f'x{i}') for i in range(DIM)])
      v = fn(x - shift, seed=seed)
      best = min(best, v)
      t = s.to_trial(uid)
      t.complete(vz.Measurement(metrics={'obj': -v}))
      d.update(CompletedTrials([t]), ActiveTrials())
  return float(best)


def main():
  out = {}
  for fname, fn in (('Sphere', bbob.Sphere), ('Rastrigin', bbob.Rastrigin)):
    for name, factory in designers().items():
      t0 = time.time()
      v = run(factory, fn)
      out[f'{fname}/{name}'] = v
      print(f'{fname:10s} {name:14s} best@100 = {v:10.4f} '
            f'({time.time()-t0:.1f}s)', flush=True)
  with open('profiles/cpu_designers.json', 'w') as f:
    json.dump(out, f, indent=2)
  print('wrote profiles/cpu_designers.json')


if __name__ == '__main__':
  main()
