"""Transfer learning: stacked residual GPs across studies.

Capability parity with vizier/_src/algorithms/designers/gp/gp_models.py
(StackedResidualGP :91, train_stacked_residual_gp :245-299) and
gp/transfer_learning.py (dof-weighted stddev combination :46-152):
a base GP trained on prior-study data predicts the current study's
labels; a top GP is trained on the residuals; predictions combine mean
= base + top and an inflated stddev that discounts the base model by
its degrees of freedom.
"""

from __future__ import annotations

import dataclasses
import math
from typing import List, Optional, Sequence, Tuple

import torch

from vizier_amd._src.gp import gp_model


@dataclasses.dataclass
class TransferPrediction:
  mean: torch.Tensor
  stddev: torch.Tensor


def combine_predictions(top_mean: torch.Tensor, top_stddev: torch.Tensor,
                        base_mean: torch.Tensor,
                        base_stddev: torch.Tensor, *,
                        num_obs_base: int,
                        num_obs_top: int) -> TransferPrediction:
  """Mean = base + top; stddev mixes the two with dof weighting.

  The base model's uncertainty is inflated when it was fit on little
  data relative to the top model (transfer_learning.py:46).
  """
  mean = base_mean + top_mean
  alpha = num_obs_base / max(num_obs_base + num_obs_top, 1)
  var = top_stddev ** 2 + alpha * base_stddev ** 2
  return TransferPrediction(mean=mean, stddev=var.clamp_min(1e-12).sqrt())


class StackedResidualGP:
  """A top GP over the residuals of a (possibly stacked) base GP."""

  def __init__(self, top: gp_model.GPPosterior,
               base: Optional['StackedResidualGP'],
               num_obs: int):
    self._top = top
    self._base = base
    self._num_obs = num_obs

  @property
  def num_obs(self) -> int:
    return self._num_obs

  def predict(self, xq: torch.Tensor) -> Tuple[torch.Tensor, torch.Tensor]:
    top_mean, top_stddev = self._top.predict(xq)
    if self._base is None:
      return top_mean, top_stddev
    base_mean, base_stddev = self._base.predict(xq)
    combined = combine_predictions(
        top_mean, top_stddev, base_mean, base_stddev,
        num_obs_base=self._base.num_obs, num_obs_top=self._num_obs)
    return combined.mean, combined.stddev


def train_stacked_gp(datasets: Sequence[Tuple[torch.Tensor, torch.Tensor]],
                     *, num_restarts: int = 4, max_iters: int = 50,
                     seed: int = 0) -> StackedResidualGP:
  """Trains a chain of residual GPs, one per dataset (oldest first).

  The last dataset is the current study; earlier ones are prior studies
  whose GPs become priors via residual stacking
  (gp_models.py:302-365's sequential train_gp).
  """
  if not datasets:
    raise ValueError('Need at least one (x, y) dataset.')
  stacked: Optional[StackedResidualGP] = None
  for i, (x, y) in enumerate(datasets):
    y = y.reshape(-1)
    if stacked is not None:
      base_mean, _ = stacked.predict(x)
      residual = y - base_mean
    else:
      residual = y
    post = gp_model.train_gp(x, residual, num_restarts=num_restarts,
                             max_iters=max_iters, seed=seed + i)
    stacked = StackedResidualGP(post, stacked, num_obs=x.shape[0])
  return stacked
