"""rocprof target: N warm train_gp refits at the headline shape only."""
import sys
import time

import torch

sys.path.insert(0, '.')
from vizier_amd._src.gp import gp_model  # noqa: E402

torch.manual_seed(0)
x = torch.rand(1000, 20, device='cuda')
y = torch.sin(x.sum(-1)).cuda()

# Cold fit once to get a warm-start raw (excluded from the profile's
# interpretation by its distinct restart count).
post = gp_model.train_gp(x, y, num_restarts=2, max_iters=12, seed=0,
                         precompute_inverse=False)
raw = post.raw
torch.cuda.synchronize()
t0 = time.perf_counter()
for i in range(10):
  gp_model.train_gp(x, y, num_restarts=2, max_iters=12, seed=i,
                    warm_start_raw=raw, precompute_inverse=False)
torch.cuda.synchronize()
print(f'warm train_gp: {(time.perf_counter()-t0)/10*1e3:.1f} ms/call')
