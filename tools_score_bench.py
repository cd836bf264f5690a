"""Measure fused posterior_score kernel vs composed (gram + rocBLAS)."""
import sys, time
import torch
sys.path.insert(0, '.')
from vizier_amd._src.gp import gp_model, acquisitions as acq_lib
from vizier_amd._src.ops import dispatch as ops

g = torch.Generator().manual_seed(0)
x = torch.rand(1000, 20, generator=g).cuda()
y = torch.sin(3 * x[:, 0]) + x[:, 1]
post = gp_model.train_gp(x, y, num_restarts=2, max_iters=15, seed=0)
ext = ops.require_ext()
xq = torch.rand(25, 20).cuda()
onehot = torch.zeros(20, dtype=torch.uint8).cuda()
amp = float(post.params.amplitude); mean_c = float(post.params.mean)
ls = post.params.lengthscales.contiguous()
alpha = post.alpha.contiguous(); kinv = post.K_inv.contiguous()
amp2 = amp * amp

def fused():
    return ext.posterior_scores(xq, post.x, ls, amp, mean_c, alpha, kinv,
                                onehot, 0, 1.8, 0.0, 0.0)

def composed():
    k = ext.gram_matern52(xq, post.x, ls, amp)        # (B, N) hand kernel
    t = k @ kinv                                       # rocBLAS
    var = (amp2 - (k * t).sum(-1)).clamp_min(1e-12)
    mu = mean_c + k @ alpha
    return mu + 1.8 * var.sqrt()

for name, fn in (('fused', fused), ('composed', composed)):
    for _ in range(20): fn()
    torch.cuda.synchronize(); t0 = time.perf_counter()
    for _ in range(200): fn()
    torch.cuda.synchronize()
    dt = (time.perf_counter() - t0) / 200
    print(f'{name}: {dt*1e6:.1f} us/call', flush=True)

a, b = fused(), composed()
# fused includes trust region; composed here does not -> compare masked
print('max |fused-composed| (TR off):',
      float((fused().cpu() - composed().cpu()).abs().max()))

def chunked():
    return ext.posterior_scores_chunked(xq, post.x, ls, amp, mean_c,
                                        alpha, kinv, onehot, 0, 1.8, 0.0,
                                        0.0)

for _ in range(20): chunked()
torch.cuda.synchronize(); t0 = time.perf_counter()
for _ in range(200): chunked()
torch.cuda.synchronize()
print(f'chunked: {(time.perf_counter()-t0)/200*1e6:.1f} us/call',
      flush=True)
print('max |chunked-composed|:',
      float((chunked().cpu() - composed().cpu()).abs().max()))
