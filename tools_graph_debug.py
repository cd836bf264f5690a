"""Debug: does the hipGraph sweep path capture, and how fast is it?"""
import sys, time
import torch
sys.path.insert(0, '.')
from vizier_amd._src.algorithms.optimizers.vectorized import (
    VectorizedOptimizerFactory)
from vizier_amd._src.algorithms.optimizers.eagle import CandidateBatch

def score(batch):
    x = batch.continuous[:, 0, :]
    return -((x - 0.7) ** 2).sum(-1)

for evals in (2500, 75000):
    fac = VectorizedOptimizerFactory(max_evaluations=evals,
                                     suggestion_batch_size=25)
    opt = fac(n_continuous=20, categorical_sizes=[], seed=0, device='cuda')
    torch.cuda.synchronize(); t0 = time.perf_counter()
    res = opt.optimize(score, count=1)
    torch.cuda.synchronize(); dt = time.perf_counter() - t0
    print(f'evals={evals} time={dt:.3f}s used_graph={opt.last_used_graph} '
          f'err={opt.last_graph_error} best={float(res.rewards[0]):.5f}',
          flush=True)

# Also with a real GP score (fused kernel), like the bench.
from vizier_amd._src.gp import gp_model, acquisitions as acq_lib
g = torch.Generator().manual_seed(0)
x = torch.rand(1000, 20, generator=g).cuda()
y = torch.sin(3 * x[:, 0]) + x[:, 1]
post = gp_model.train_gp(x, y, num_restarts=2, max_iters=15, seed=0)
scoring = acq_lib.ScoringFunction(post, acq_lib.UCB(1.8),
                                  acq_lib.TrustRegion(post.x))
def gp_score(batch):
    return scoring(batch.continuous[:, 0, :])
fac = VectorizedOptimizerFactory(max_evaluations=75000,
                                 suggestion_batch_size=25)
opt = fac(n_continuous=20, categorical_sizes=[], seed=0, device='cuda')
torch.cuda.synchronize(); t0 = time.perf_counter()
res = opt.optimize(gp_score, count=1)
torch.cuda.synchronize(); dt = time.perf_counter() - t0
print(f'GP-score sweep 75k: {dt:.3f}s used_graph={opt.last_used_graph} '
      f'err={opt.last_graph_error}', flush=True)
