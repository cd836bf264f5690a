"""Per-kernel timing of the headline sweep (hipGraph path, N=1000)."""
import sys, time
import numpy as np
import torch
sys.path.insert(0, '.')
from vizier_amd._src.algorithms.core.abstractions import ActiveTrials, CompletedTrials
from vizier_amd._src.algorithms.designers.gp_bandit import GPBanditConfig, VizierGPBandit
import bench as B

problem = B.make_problem()
designer = VizierGPBandit(problem, GPBanditConfig(
    max_evaluations=75000, suggestion_batch_size=25, device='cuda'), seed=0)
rng = np.random.default_rng(0)
trials = []
for uid in range(1, 1001):
  params = {f'x{i}': float(v) for i, v in enumerate(rng.uniform(-5, 5, 20))}
  trials.append(B.trial_from(params, uid))
designer.update(CompletedTrials(trials), ActiveTrials())
designer.suggest(1)  # warm
torch.cuda.synchronize()
t0 = time.perf_counter()
designer.suggest(1)
torch.cuda.synchronize()
print(f'suggest: {(time.perf_counter()-t0)*1e3:.1f} ms', flush=True)
