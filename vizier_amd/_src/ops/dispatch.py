"""Op dispatch: hand-written CDNA4 HIP kernels on GPU, PyTorch on CPU.

Policy: on a GPU (ROCm) machine the HIP extension MUST be present and is
always used for GPU tensors — if it fails to import, GPU ops raise
loudly rather than silently falling back to eager PyTorch. On CPU the
PyTorch reference implementations run (they are also the numerics oracle
for the kernels, compared in tests/test_gpu_ops.py).
"""

from __future__ import annotations

from typing import Optional

import torch

from vizier_amd._src.gp import matern as _torch_matern

_EXT = None
_EXT_ERROR: Optional[str] = None

ACQ_CODES = {'ucb': 0, 'lcb': 1, 'ei': 2, 'pi': 3, 'mean': 4, 'stddev': 5}


def _load_extension():
  global _EXT, _EXT_ERROR
  if _EXT is not None or _EXT_ERROR is not None:
    return _EXT
  try:
    import vizier_amd_hip  # built in-tree by setup.py / __graft_entry__
    _EXT = vizier_amd_hip
  except ImportError as e:  # pragma: no cover - GPU box only
    _EXT_ERROR = str(e)
  return _EXT


def extension_available() -> bool:
  return _load_extension() is not None


def require_ext():
  ext = _load_extension()
  if ext is None:
    raise RuntimeError(
        'vizier_amd_hip extension is required for GPU tensors but could '
        f'not be imported: {_EXT_ERROR}. Build it with '
        '`python setup.py build_ext --inplace` (or __graft_entry__.build).')
  return ext


def gram_matern52(x1: torch.Tensor, x2: Optional[torch.Tensor],
                  lengthscales: torch.Tensor,
                  amplitude: torch.Tensor) -> torch.Tensor:
  """Matern-5/2 ARD Gram / cross-Gram matrix."""
  if x1.is_cuda and x1.dtype == torch.float32:
    ext = require_ext()
    x2t = x1 if x2 is None else x2
    return ext.gram_matern52(x1.contiguous(), x2t.contiguous(),
                             lengthscales.contiguous(), float(amplitude))
  return _torch_matern.gram_matern52(x1, x2, lengthscales, amplitude)


def fused_posterior_scores(xq: torch.Tensor, x: torch.Tensor,
                           lengthscales: torch.Tensor, amplitude: float,
                           mean_c: float, alpha: torch.Tensor,
                           k_inv: torch.Tensor, onehot_u8: torch.Tensor,
                           acq: str, coef: float, best_value: float,
                           tr_radius: float) -> torch.Tensor:
  """One-launch GP posterior + acquisition + trust region (GPU only)."""
  ext = require_ext()
  return ext.posterior_scores(
      xq.contiguous(), x.contiguous(), lengthscales.contiguous(),
      float(amplitude), float(mean_c), alpha.contiguous(),
      k_inv.contiguous(), onehot_u8.contiguous(), ACQ_CODES[acq],
      float(coef), float(best_value), float(tr_radius))
