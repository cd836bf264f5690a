"""Phase timing for one suggest(): ARD fit vs Eagle sweep vs pieces."""
import time, sys
import numpy as np
import torch
sys.path.insert(0, '.')
from vizier_amd import pyvizier as vz
from vizier_amd._src.algorithms.core.abstractions import ActiveTrials, CompletedTrials
from vizier_amd._src.algorithms.designers.gp_bandit import GPBanditConfig, VizierGPBandit
import bench as B

problem = B.make_problem()
cfg = GPBanditConfig(max_evaluations=75000, suggestion_batch_size=25,
                     ard_restarts=4, ard_max_iters=50, device='cuda')
designer = VizierGPBandit(problem, cfg, seed=0)
rng = np.random.default_rng(0)
trials = []
for uid in range(1, 1001):
    params = {f'x{i}': float(v) for i, v in enumerate(rng.uniform(-5, 5, 20))}
    trials.append(B.trial_from(params, uid))
designer.update(CompletedTrials(trials), ActiveTrials())

def t(fn):
    torch.cuda.synchronize(); t0=time.perf_counter(); out=fn(); torch.cuda.synchronize()
    return out, time.perf_counter()-t0

# warm up once fully
_, warm = t(lambda: designer.suggest(1))
print('warm suggest total:', round(warm,3), 's', flush=True)

# new trial -> forces refit
designer.update(CompletedTrials([B.trial_from({f'x{i}': 0.1*i for i in range(20)}, 1001)]), ActiveTrials())
_, fit_t = t(designer._fit)
print('fit:', round(fit_t,3), 's', flush=True)
_, sweep_t = t(lambda: designer._gp_suggestions(1))
print('sweep (incl cached fit):', round(sweep_t,3), 's', flush=True)

# micro: single eagle iteration cost
from vizier_amd._src.algorithms.optimizers.vectorized import VectorizedOptimizerFactory
score_fn, _ = designer._score_factory(1)
fac = VectorizedOptimizerFactory(max_evaluations=2500, suggestion_batch_size=25)
opt = fac(n_continuous=20, categorical_sizes=[], seed=0, device='cuda')
_, t100 = t(lambda: opt.optimize(score_fn, count=1))
print('100 iterations of sweep:', round(t100,3), 's =>', round(t100/100*1000,3), 'ms/iter', flush=True)
