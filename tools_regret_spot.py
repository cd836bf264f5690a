"""Quick regret spot-check after fit-path changes (subset of
tools_regret_bench.py: Sphere + Rastrigin, gp_bandit_ucb, 3 seeds)."""
import importlib.util
import json
import sys
import time

sys.path.insert(0, '.')
spec = importlib.util.spec_from_file_location('rb', 'tools_regret_bench.py')
rb = importlib.util.module_from_spec(spec)
spec.loader.exec_module(rb)

import torch  # noqa: E402
from vizier_amd._src.benchmarks.experimenters.synthetic import bbob  # noqa: E402
from vizier_amd._src.algorithms.designers.gp_bandit import (  # noqa: E402
    GPBanditConfig,
    VizierGPBandit,
)

device = 'cuda' if torch.cuda.is_available() else 'cpu'


def gp_bandit(problem, seed):
  return VizierGPBandit(problem, GPBanditConfig(
      max_evaluations=10000, device=device), seed=seed)


def gp_ucb_pe(problem, seed):
  from vizier_amd._src.algorithms.designers.gp_ucb_pe import (
      UCBPEConfig,
      VizierGPUCBPEBandit,
  )
  return VizierGPUCBPEBandit(problem, UCBPEConfig(device=device),
                             seed=seed)


FULL = len(sys.argv) > 1 and sys.argv[1] == 'full'
functions = [('Sphere', bbob.Sphere), ('Rastrigin', bbob.Rastrigin)]
algos = [('gp_bandit_ucb', gp_bandit)]
seeds = 3
if FULL:
  functions += [('SharpRidge', bbob.SharpRidge),
                ('Rosenbrock', bbob.Rosenbrock)]
  algos += [('gp_ucb_pe', gp_ucb_pe)]
  seeds = 2

out = {}
for fname, fn in functions:
  for aname, factory in algos:
    bests = []
    for seed in range(seeds):
      t0 = time.time()
      best = rb.run(factory, fn, 20, 100, seed)
      bests.append(best)
      print(f'{fname} {aname} seed={seed} best@100={best:.4f} '
            f'({time.time()-t0:.1f}s)', flush=True)
    out[f'{fname}/{aname}'] = bests
print(json.dumps(out))
