"""GP-UCB-PE (DEFAULT algorithm) suggest timing, 20D N=1000, batch 4."""
import sys, time
import numpy as np
import torch
sys.path.insert(0, '.')
from vizier_amd import pyvizier as vz
from vizier_amd._src.algorithms.core.abstractions import ActiveTrials, CompletedTrials
from vizier_amd._src.algorithms.designers.gp_ucb_pe import UCBPEConfig, VizierGPUCBPEBandit

problem = vz.ProblemStatement()
for i in range(20):
  problem.search_space.root.add_float_param(f'x{i}', -5.0, 5.0)
problem.metric_information.append(vz.MetricInformation(
    name='obj', goal=vz.ObjectiveMetricGoal.MAXIMIZE))
d = VizierGPUCBPEBandit(problem, UCBPEConfig(device='cuda'), seed=0)
rng = np.random.default_rng(0)
trials = []
for uid in range(1, 1001):
  x = rng.uniform(-5, 5, 20)
  t = vz.Trial({f'x{i}': float(x[i]) for i in range(20)}, id=uid)
  t.complete(vz.Measurement(metrics={'obj': float(-(x * x).sum())}))
  trials.append(t)
d.update(CompletedTrials(trials), ActiveTrials())
d.suggest(4)  # warmup (fit + captures)
torch.cuda.synchronize()
t0 = time.perf_counter()
out = d.suggest(4)
torch.cuda.synchronize()
dt = time.perf_counter() - t0
kinds = [s.metadata.abs_ns(('gp_ucb_pe',))['acquisition'] for s in out]
print(f'suggest(4): {dt*1e3:.1f} ms, phases={kinds}', flush=True)
