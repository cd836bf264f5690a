"""Does ard_warm_iters=8 preserve regret quality? (bench fit is ~50%
of suggest; iterations are its linear knob)."""
import sys
sys.path.insert(0, '.')
import json
import numpy as np
import tools_regret_bench as rb
from vizier_amd._src.algorithms.designers.gp_bandit import (
    GPBanditConfig, VizierGPBandit)
from vizier_amd._src.benchmarks.experimenters.synthetic import bbob

def gp(warm_iters):
  def factory(problem, seed):
    return VizierGPBandit(problem, GPBanditConfig(
        max_evaluations=10000, ard_restarts=4, ard_max_iters=50,
        ard_warm_iters=warm_iters, device='cuda'), seed=seed)
  return factory

out = {}
for wi in (12, 8, 6):
  for fname, fn in (('Sphere', bbob.Sphere), ('SharpRidge', bbob.SharpRidge),
                    ('Rastrigin', bbob.Rastrigin)):
    v = rb.run(gp(wi), fn, 20, 100, seed=0)
    out[f'{fname}/warm{wi}'] = v
    print(f'{fname} warm_iters={wi}: best@100 = {v:.4f}', flush=True)
print(json.dumps(out))
