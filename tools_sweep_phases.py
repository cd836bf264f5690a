"""Reads the megakernel's per-phase cycle accumulators after a sweep."""
import sys
import time

import numpy as np
import torch

sys.path.insert(0, '.')
from vizier_amd._src.algorithms.core.abstractions import (  # noqa: E402
    ActiveTrials,
    CompletedTrials,
)
from vizier_amd._src.algorithms.designers.gp_bandit import (  # noqa: E402
    GPBanditConfig,
    VizierGPBandit,
)
from vizier_amd._src.algorithms.optimizers.vectorized import (  # noqa: E402
    VectorizedOptimizerFactory,
)
import bench as B  # noqa: E402

problem = B.make_problem()
designer = VizierGPBandit(problem, GPBanditConfig(
    max_evaluations=75000, suggestion_batch_size=25, device='cuda'),
    seed=0)
rng = np.random.default_rng(0)
trials = [B.trial_from({f'x{i}': float(v)
                        for i, v in enumerate(rng.uniform(-5, 5, 20))},
                       uid) for uid in range(1, 1001)]
designer.update(CompletedTrials(trials), ActiveTrials())
designer.suggest(1)
score_fn, _ = designer._score_factory(1)
fac = VectorizedOptimizerFactory(max_evaluations=75000,
                                 suggestion_batch_size=25)
opt = fac(n_continuous=20, categorical_sizes=[], seed=0, device='cuda')
opt.optimize(score_fn, count=1)  # warm
strat = opt.strategy
base = strat._iter_t[2:12].clone()
torch.cuda.synchronize()
t0 = time.perf_counter()
opt.optimize(score_fn, count=1)
torch.cuda.synchronize()
wall = time.perf_counter() - t0
delta = (strat._iter_t[2:12] - base).cpu().numpy().astype(float)
names = ['A', 'barA', 'B', 'barB', 'B2', 'barB2', 'C', 'barC',
         'B.stage', 'B.main']
total = delta[:8].sum()
iters = 3000 - (strat.pool_size // strat.batch_size + 2)
print(f'sweep wall: {wall*1e3:.1f} ms over ~{iters} megakernel iters '
      f'({wall/3000*1e6:.1f} us/iter incl. eager head)')
for nm, c in zip(names, delta):
  print(f'  {nm:6s} {c/total*100:5.1f}%  {c/iters:10.0f} cyc/iter')
