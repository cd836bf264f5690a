"""Config-4 (50D, N=10000) phase breakdown on one MI355X.

Times, separately: initial cold ARD fit, warm refit, posterior cache
build, and the Eagle sweep per-iteration cost — with the GEMM quadform
path (default) vs the per-candidate streaming kernel (old path,
VIZIER_AMD_PS_GEMM_N huge). Writes gpurun_out/config4_phases.json.
"""

import json
import os
import sys
import time

import numpy as np
import torch

sys.path.insert(0, '.')
from vizier_amd import pyvizier as vz
from vizier_amd._src.algorithms.core.abstractions import (
    ActiveTrials,
    CompletedTrials,
)
from vizier_amd._src.algorithms.designers.gp_bandit import (
    GPBanditConfig,
    VizierGPBandit,
)

DIM = 50
N = 10000


def make_designer(max_evals):
  p = vz.ProblemStatement()
  for i in range(DIM):
    p.search_space.root.add_float_param(f'x{i}', -5.0, 5.0)
  p.metric_information.append(vz.MetricInformation(
      name='obj', goal=vz.ObjectiveMetricGoal.MAXIMIZE))
  d = VizierGPBandit(p, GPBanditConfig(
      max_evaluations=max_evals, device='cuda'), seed=0)
  rng = np.random.default_rng(0)
  trials = []
  for uid in range(1, N + 1):
    x = rng.uniform(-5, 5, DIM)
    t = vz.Trial({f'x{i}': float(x[i]) for i in range(DIM)}, id=uid)
    t.complete(vz.Measurement(metrics={'obj': float(-(x * x).sum() / DIM)}))
    trials.append(t)
  d.update(CompletedTrials(trials), ActiveTrials())
  return d


def t_sync(fn):
  torch.cuda.synchronize()
  t0 = time.perf_counter()
  out = fn()
  torch.cuda.synchronize()
  return time.perf_counter() - t0, out


def main():
  out = {'gemm_n_threshold': os.environ.get('VIZIER_AMD_PS_GEMM_N',
                                            'default(4096)')}

  # Small sweep budget so fit phases are visible: 2000 iters x 25.
  d = make_designer(50_000)
  dt, _ = t_sync(lambda: d._fit())
  out['cold_fit_s'] = dt
  print(f'cold ARD fit (4 restarts x 50 it): {dt:.2f} s', flush=True)

  # Warm refit: add one trial, refit.
  sugg = d._seed_suggestions(1)
  t = sugg[0].to_trial(N + 1)
  t.complete(vz.Measurement(metrics={'obj': 0.0}))
  d.update(CompletedTrials([t]), ActiveTrials())
  dt, _ = t_sync(lambda: d._fit())
  out['warm_refit_s'] = dt
  print(f'warm ARD refit (2 restarts x 12 it): {dt:.2f} s', flush=True)

  # Sweep: first suggest pays graph capture; second is steady state.
  dt, _ = t_sync(lambda: d.suggest(1))
  out['suggest_with_capture_s'] = dt
  print(f'suggest #1 (incl capture, 2000 iters): {dt:.2f} s', flush=True)
  dt, _ = t_sync(lambda: d.suggest(1))
  out['suggest_steady_s'] = dt
  out['sweep_ms_per_iter'] = dt / 2000 * 1e3
  print(f'suggest #2 (steady, 2000 iters): {dt:.2f} s '
        f'=> {dt / 2000 * 1e3:.3f} ms/iter', flush=True)

  path = 'gpurun_out/config4_phases.json'
  old = json.load(open(path)) if os.path.exists(path) else {}
  key = ('gemm' if out['gemm_n_threshold'].startswith('default') or
         int(out['gemm_n_threshold']) <= N else 'stream')
  old[key] = out
  os.makedirs('gpurun_out', exist_ok=True)
  with open(path, 'w') as f:
    json.dump(old, f, indent=2)
  print(json.dumps(old))


if __name__ == '__main__':
  main()
