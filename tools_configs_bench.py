"""Measures BASELINE.json's milestone configs 2-5 on ONE MI355X.

  2: GP-UCB 8D, N=500, bf16 candidate grams
  3: GP-EI (qEI) 20D, N=2000, suggest count=8
  4: GP-Bandit 50D, N=10000, 1.25M-candidate sweep (the per-GPU slice
     of the 10M-candidate DP=8 config; the driver runs the multi-GPU
     scaling itself)
  5: Multi-objective HV-scalarized UCB + trust region, 30D, fp8 grams

Each reports suggest() wall-clock (mean over a few calls, first call
excluded: it pays hipGraph/megakernel warmup).
"""

import json
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


def make_problem(dim, metrics=('obj',)):
  p = vz.ProblemStatement()
  for i in range(dim):
    p.search_space.root.add_float_param(f'x{i}', -5.0, 5.0)
  p.metric_information = vz.MetricsConfig([
      vz.MetricInformation(name=m, goal=vz.ObjectiveMetricGoal.MAXIMIZE)
      for m in metrics])
  return p


def preload(designer, dim, n_trials, metrics=('obj',), seed=0):
  rng = np.random.default_rng(seed)
  trials = []
  for uid in range(1, n_trials + 1):
    x = rng.uniform(-5, 5, dim)
    t = vz.Trial({f'x{i}': float(x[i]) for i in range(dim)}, id=uid)
    vals = {m: float(-((x - k) ** 2).sum() / dim)
            for k, m in enumerate(metrics)}
    t.complete(vz.Measurement(metrics=vals))
    trials.append(t)
  designer.update(CompletedTrials(trials), ActiveTrials())


def timed_suggest(designer, count=1, reps=3):
  designer.suggest(count)  # warmup (fit + graph capture)
  torch.cuda.synchronize()
  times = []
  uid = 10_000_000
  for _ in range(reps):
    t0 = time.perf_counter()
    sugg = designer.suggest(count)
    torch.cuda.synchronize()
    times.append(time.perf_counter() - t0)
    # Complete one so the fit isn't trivially cached.
    uid += 1
    t = sugg[0].to_trial(uid)
    t.complete(vz.Measurement(metrics={
        m.name: 0.0 for m in designer._problem.metric_information}))
    designer.update(CompletedTrials([t]), ActiveTrials())
  return float(np.mean(times) * 1e3)


def run_one(which: str) -> None:
  main(only=which)


def main(only=None):
  out = {}

  # Config 2: 8D, N=500, bf16 grams.
  for dtype in (('fp32', 'bf16') if only in (None, '2') else ()):
    d = VizierGPBandit(make_problem(8), GPBanditConfig(
        max_evaluations=75000, device='cuda',
        scorer_gram_dtype=dtype), seed=0)
    preload(d, 8, 500)
    ms = timed_suggest(d)
    out[f'config2_8d_n500_{dtype}'] = ms
    print(f'config 2 (8D N=500, {dtype}): {ms:.1f} ms/suggest',
          flush=True)

  # Config 3: 20D, N=2000, qEI count=8.
  if only in (None, '3'):
    _config3(out)

  # Config 4: 50D, N=10000, 1.25M evals (per-GPU slice of 10M DP=8).
  if only in (None, '4'):
    _config4(out)

  # Config 5: 30D MO + trust region, fp8 grams.
  if only in (None, '5'):
    _config5(out)

  print(json.dumps(out))
  import os
  path = 'gpurun_out/configs_bench.json'
  old = json.load(open(path)) if os.path.exists(path) else {}
  old.update(out)
  with open(path, 'w') as f:
    json.dump(old, f, indent=2)


def _config3(out):
  d = VizierGPBandit(make_problem(20), GPBanditConfig(
      max_evaluations=75000, acquisition='qei', device='cuda'), seed=0)
  preload(d, 20, 2000)
  ms = timed_suggest(d, count=8)
  out['config3_20d_n2000_qei8'] = ms
  print(f'config 3 (20D N=2000 qEI q=8): {ms:.1f} ms/suggest(8)',
        flush=True)


def _config4(out):
  d = VizierGPBandit(make_problem(50), GPBanditConfig(
      max_evaluations=1_250_000, device='cuda'), seed=0)
  preload(d, 50, 10000)
  ms = timed_suggest(d, reps=2)
  out['config4_50d_n10000_1p25M'] = ms
  print(f'config 4 (50D N=10000, 1.25M evals): {ms:.1f} ms/suggest',
        flush=True)


def _config5(out):
  for dtype in ('fp32', 'fp8'):
    d = VizierGPBandit(make_problem(30, metrics=('f1', 'f2')),
                       GPBanditConfig(max_evaluations=75000,
                                      device='cuda',
                                      scorer_gram_dtype=dtype), seed=0)
    preload(d, 30, 1000, metrics=('f1', 'f2'))
    ms = timed_suggest(d)
    out[f'config5_30d_mo_{dtype}'] = ms
    print(f'config 5 (30D MO N=1000, {dtype}): {ms:.1f} ms/suggest',
          flush=True)
    del d
    torch.cuda.empty_cache()


if __name__ == '__main__':
  import sys as _sys
  main(only=_sys.argv[1] if len(_sys.argv) > 1 else None)
