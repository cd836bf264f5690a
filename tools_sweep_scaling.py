"""Megakernel sweep us/iter vs N — separates fixed per-iteration
overhead (grid barriers, phase latency) from N-scaled work (k-vector,
quadform)."""
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

for n in (125, 250, 500, 1000):
  problem = B.make_problem()
  designer = VizierGPBandit(problem, GPBanditConfig(
      max_evaluations=75000, suggestion_batch_size=25, device='cuda'),
      seed=0)
  rng = np.random.default_rng(0)
  trials = []
  for uid in range(1, n + 1):
    params = {f'x{i}': float(v)
              for i, v in enumerate(rng.uniform(-5, 5, 20))}
    trials.append(B.trial_from(params, uid))
  designer.update(CompletedTrials(trials), ActiveTrials())
  designer.suggest(1)  # warm: fit + capture
  score_fn, _ = designer._score_factory(1)
  fac = VectorizedOptimizerFactory(max_evaluations=75000,
                                   suggestion_batch_size=25)
  opt = fac(n_continuous=20, categorical_sizes=[], seed=0,
            device='cuda')
  opt.optimize(score_fn, count=1)  # warm
  torch.cuda.synchronize()
  t0 = time.perf_counter()
  opt.optimize(score_fn, count=1)
  torch.cuda.synchronize()
  dt = time.perf_counter() - t0
  print(f'N={n:5d}: sweep {dt*1e3:7.1f} ms '
        f'({dt/3000*1e6:5.1f} us/iter) mega={opt.last_used_megakernel}',
        flush=True)
