"""Map torch.linalg.cholesky_ex crashes on ROCm by (batch, N) and
backend (default vs forced hipSOLVER/MAGMA). Each case in a subprocess."""
import subprocess, sys
sys.path.insert(0, '.')

def case(r, n, backend):
  import torch
  if backend != 'default':
    torch.backends.cuda.preferred_linalg_library(backend)
  g = torch.Generator().manual_seed(0)
  a = torch.randn(r, n, n, generator=g).cuda()
  k = a @ a.mT + n * torch.eye(n, device='cuda')
  L, info = torch.linalg.cholesky_ex(k)
  torch.cuda.synchronize()
  print('OK', float(L.diagonal(dim1=-2, dim2=-1).sum()))

if __name__ == '__main__':
  if len(sys.argv) > 1:
    case(int(sys.argv[1]), int(sys.argv[2]), sys.argv[3])
  else:
    import itertools
    ns = [128, 160, 200, 255, 256, 257, 288, 300, 320, 384, 448,
          511, 512, 513, 576, 640, 768, 900, 1000, 1024, 1500]
    for backend in ('default',):
      for (r, n) in itertools.chain(((4, n) for n in ns),
                                    ((2, n) for n in (300, 400, 500))):
        p = subprocess.run(['timeout', '120', sys.executable, __file__,
                            str(r), str(n), backend],
                           capture_output=True, text=True)
        out = p.stdout.strip().splitlines()
        msg = out[-1] if out else p.stderr.strip().splitlines()[-1][:60] if p.stderr.strip() else '?'
        print(f'{backend:9s} R={r:3d} N={n:5d}: rc={p.returncode} {msg}',
              flush=True)
