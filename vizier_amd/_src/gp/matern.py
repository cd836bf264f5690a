"""Matern-5/2 ARD kernel (PyTorch reference implementation).

This is the numeric spec for the CDNA4 HIP Gram kernels (see
vizier_amd/_src/ops): the GPU path computes the same quantity with a
fused LDS-tiled pairwise-distance + kernel evaluation. Mirrors the math
of the reference's tfpk.MaternFiveHalves + FeatureScaledWithCategorical
(vizier/_src/jax/models/tuned_gp_models.py:170,195), with categoricals
one-hot-embedded by the converter so all features are continuous.
"""

from __future__ import annotations

import math
from typing import Optional

import torch

_SQRT5 = math.sqrt(5.0)


def matern52(r: torch.Tensor) -> torch.Tensor:
  """Matern-5/2 correlation as a function of scaled distance r >= 0."""
  sr = _SQRT5 * r
  return (1.0 + sr + sr * sr / 3.0) * torch.exp(-sr)


def pairwise_sqdist(x1: torch.Tensor, x2: torch.Tensor) -> torch.Tensor:
  """Squared euclidean distances between row sets, clamped at 0.

  Uses the |a|^2 + |b|^2 - 2ab GEMM form so the GPU path maps onto MFMA.
  Supports batched inputs (..., N, D) x (..., M, D) -> (..., N, M).
  """
  n1 = (x1 * x1).sum(-1, keepdim=True)          # (..., N, 1)
  n2 = (x2 * x2).sum(-1, keepdim=True)          # (..., M, 1)
  d2 = n1 + n2.transpose(-1, -2) - 2.0 * (x1 @ x2.transpose(-1, -2))
  return d2.clamp_min_(0.0)


def gram_matern52(x1: torch.Tensor, x2: Optional[torch.Tensor],
                  lengthscales: torch.Tensor,
                  amplitude: torch.Tensor) -> torch.Tensor:
  """K[i,j] = amplitude^2 * m52(||(x1_i - x2_j) / lengthscales||).

  Args:
    x1: (..., N, D) features.
    x2: (..., M, D) features, or None for x2 = x1.
    lengthscales: (..., D) or (..., 1, D) per-dimension ARD lengthscales.
    amplitude: scalar or (...) batch of amplitudes.
  """
  if lengthscales.dim() == x1.dim() - 1:
    lengthscales = lengthscales.unsqueeze(-2)
  z1 = x1 / lengthscales
  z2 = z1 if x2 is None else x2 / lengthscales
  # Epsilon below fp32 resolution: sqrt(0) has a NaN gradient, which
  # would silently freeze an entire L-BFGS restart when two rows
  # coincide (the diagonal does, whenever the GEMM form cancels
  # exactly). matern52(1e-9) == 1.0f exactly, so values are unchanged.
  r = pairwise_sqdist(z1, z2).clamp_min_(1e-18).sqrt()
  k = matern52(r)
  amp2 = (amplitude * amplitude)
  if amp2.dim() > 0:
    amp2 = amp2.reshape(amp2.shape + (1, 1))
  return amp2 * k
