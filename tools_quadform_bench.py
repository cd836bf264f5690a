"""A/B: custom split-K quadform kernel vs rocBLAS GEMM at config-4 shape.

Usage (on GPU box):
  python tools_quadform_bench.py custom
  VIZIER_AMD_QUADFORM=gemm python tools_quadform_bench.py gemm
Also checks numerics vs a float64 torch oracle.
"""

import json
import os
import sys
import time

import numpy as np
import torch

sys.path.insert(0, '.')
import vizier_amd_hip as ext  # noqa: E402

MODE = sys.argv[1] if len(sys.argv) > 1 else 'custom'


def main():
  torch.manual_seed(0)
  out = {}
  for n in (4096, 10000):
    b, d = 25, 50
    xq = torch.rand(b, d, device='cuda')
    x = torch.rand(n, d, device='cuda')
    ls = torch.full((d,), 2.0, device='cuda')
    alpha = torch.randn(n, device='cuda') / n
    A = torch.randn(n, 64, device='cuda', dtype=torch.float64)
    kinv64 = (A @ A.T / 64 + torch.eye(n, device='cuda',
                                       dtype=torch.float64))
    kinv = kinv64.float().contiguous()
    onehot = torch.zeros(d, dtype=torch.uint8, device='cuda')

    def run():
      return ext.posterior_scores_chunked(
          xq, x, ls, 1.0, 0.0, alpha, kinv, onehot, 0, 1.8, 0.0, 0.0)

    got = run()
    # fp64 oracle of the same pipeline.
    z1 = (xq / ls).double()
    z2 = (x / ls).double()
    d2 = ((z1 ** 2).sum(1)[:, None] + (z2 ** 2).sum(1)[None, :]
          - 2 * z1 @ z2.T).clamp_min(0)
    r = d2.sqrt()
    sr = (5.0 ** 0.5) * r
    k = (1 + sr + sr * sr / 3) * torch.exp(-sr)
    mu = k @ alpha.double()
    quad = (k * (k @ kinv64)).sum(-1)
    want = (mu + 1.8 * (1.0 - quad).clamp_min(1e-12).sqrt()).float()
    err = float((got - want).abs().max())
    out[f'n{n}_max_err_vs_fp64'] = err

    # Timing: 50 calls.
    for _ in range(5):
      run()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(50):
      run()
    torch.cuda.synchronize()
    ms = (time.perf_counter() - t0) / 50 * 1e3
    out[f'n{n}_ms_per_call'] = ms
    print(f'[{MODE}] n={n}: {ms:.3f} ms/call, max err {err:.2e}',
          flush=True)

  path = 'gpurun_out/quadform_ab.json'
  old = json.load(open(path)) if os.path.exists(path) else {}
  old[MODE] = out
  os.makedirs('gpurun_out', exist_ok=True)
  with open(path, 'w') as f:
    json.dump(old, f, indent=2)


if __name__ == '__main__':
  main()
