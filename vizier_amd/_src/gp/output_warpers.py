"""Label pre-processing warpers for GP training.

Capability parity with vizier/_src/algorithms/designers/gp/output_warpers.py
(HalfRank :289, LogWarper :381, InfeasibleWarper :419, ZScore :496,
Normalize :530, create_default_warper :185). Same transforms, vectorized
NumPy implementation. All warpers take/return (N, 1) arrays; NaN encodes
infeasible labels.
"""

from __future__ import annotations

import abc
from typing import List, Optional, Sequence, Tuple

import numpy as np
from scipy import stats


def _validate(labels: np.ndarray) -> np.ndarray:
  labels = np.asarray(labels, dtype=np.float64)
  if labels.ndim == 1:
    labels = labels[:, None]
  if labels.ndim != 2 or labels.shape[1] != 1:
    raise ValueError(f'Labels must be (N, 1); got {labels.shape}')
  return labels.copy()


class OutputWarper(abc.ABC):

  @abc.abstractmethod
  def warp(self, labels_arr: np.ndarray) -> np.ndarray:
    ...

  def unwarp(self, labels_arr: np.ndarray) -> np.ndarray:
    raise NotImplementedError

  def __call__(self, labels_arr: np.ndarray) -> np.ndarray:
    return self.warp(labels_arr)


class OutputWarperPipeline(OutputWarper):

  def __init__(self, warpers: Sequence[OutputWarper] = ()):
    self.warpers = list(warpers)

  def warp(self, labels_arr: np.ndarray) -> np.ndarray:
    labels_arr = _validate(labels_arr)
    for w in self.warpers:
      labels_arr = w.warp(labels_arr)
    return labels_arr

  def unwarp(self, labels_arr: np.ndarray) -> np.ndarray:
    for w in reversed(self.warpers):
      labels_arr = w.unwarp(labels_arr)
    return labels_arr


class HalfRankComponent(OutputWarper):
  """Maps below-median labels to Gaussian quantiles (good half untouched)."""

  def _estimate_std_of_good_half(self, unique_labels: np.ndarray,
                                 threshold: float) -> float:
    good = unique_labels[unique_labels >= threshold]
    std = np.sqrt(((good - threshold) ** 2).sum() / max(len(good), 1))
    if std > 0:
      return std
    std = np.sqrt(((unique_labels - threshold) ** 2).sum()
                  / len(unique_labels))
    if np.isfinite(std) and std > 0:
      return std
    return float(np.abs(unique_labels - threshold).sum()
                 / len(unique_labels)) or 1.0

  def warp(self, labels_arr: np.ndarray) -> np.ndarray:
    labels_arr = _validate(labels_arr)
    if labels_arr.size <= 1:
      return labels_arr
    flat = labels_arr.flatten()
    finite = np.isfinite(flat)
    if finite.sum() <= 1:
      return labels_arr
    median = np.nanmedian(flat[finite])
    unique_labels = np.unique(flat[finite])
    ranks = stats.rankdata(flat, method='dense', nan_policy='omit')
    median_idx = unique_labels.searchsorted(median, 'left')
    denominator = median_idx + 0.5 * float(
        median_idx < len(unique_labels) and
        unique_labels[median_idx] == median)
    if denominator <= 0:
      return labels_arr
    std = self._estimate_std_of_good_half(unique_labels, median)
    below = finite & (flat < median)
    quantiles = 0.5 * (ranks[below] - 0.5) / denominator
    flat[below] = stats.norm.ppf(quantiles) * std + median
    return flat[:, None]


class LogWarperComponent(OutputWarper):
  """Compresses the bad tail: y -> 0.5 - log1p(norm_diff*(c-1))/log(c)."""

  def __init__(self, offset: float = 1.5):
    if offset <= 0:
      raise ValueError('offset must be positive')
    self.offset = offset
    self._labels_min: Optional[float] = None
    self._labels_max: Optional[float] = None

  def warp(self, labels_arr: np.ndarray) -> np.ndarray:
    labels_arr = _validate(labels_arr)
    flat = labels_arr.flatten()
    if np.all(np.isnan(flat)):
      return labels_arr
    self._labels_min = float(np.nanmin(flat))
    self._labels_max = float(np.nanmax(flat))
    finite = np.isfinite(flat)
    span = self._labels_max - self._labels_min
    if span == 0:
      flat[finite] = 0.5 - (np.log1p(0.0) / np.log(self.offset))
      return flat[:, None]
    norm_diff = (self._labels_max - flat[finite]) / span
    flat[finite] = 0.5 - (np.log1p(norm_diff * (self.offset - 1))
                          / np.log(self.offset))
    return flat[:, None]

  def unwarp(self, labels_arr: np.ndarray) -> np.ndarray:
    if self._labels_max is None:
      raise ValueError('warp() must be called before unwarp().')
    flat = np.asarray(labels_arr, dtype=np.float64).flatten()
    span = self._labels_max - self._labels_min
    flat = self._labels_max - (np.exp(np.log(self.offset) * (0.5 - flat))
                               - 1) * span / (self.offset - 1)
    return flat[:, None]


class InfeasibleWarperComponent(OutputWarper):
  """Replaces NaNs with a clearly-bad value and centers the labels."""

  def __init__(self):
    self._shift: Optional[float] = None

  def warp(self, labels_arr: np.ndarray) -> np.ndarray:
    labels_arr = _validate(labels_arr)
    flat = labels_arr.flatten()
    if np.all(np.isnan(flat)):
      self._shift = np.nan
      return np.zeros_like(flat)[:, None]
    labels_range = np.nanmax(flat) - np.nanmin(flat)
    warped_bad = np.nanmin(flat) - (0.5 * labels_range + 1)
    num_feasible = flat.size - np.isnan(flat).sum()
    p_feasible = (0.5 + num_feasible) / (1 + flat.size)
    self._shift = (-np.nanmean(flat) * p_feasible
                   - warped_bad * (1 - p_feasible))
    nan_mask = np.isnan(flat)
    flat[nan_mask] = warped_bad
    flat[~nan_mask] += self._shift
    return flat[:, None]

  def unwarp(self, labels_arr: np.ndarray) -> np.ndarray:
    if self._shift is None:
      raise ValueError('warp() must be called before unwarp().')
    return np.asarray(labels_arr) - self._shift


class ZScoreLabels(OutputWarper):

  def warp(self, labels_arr: np.ndarray) -> np.ndarray:
    labels_arr = _validate(labels_arr)
    flat = labels_arr.flatten()
    if np.all(np.isnan(flat)):
      raise ValueError('Labels need at least one non-NaN entry.')
    finite = np.isfinite(flat)
    std = np.nanstd(flat[finite])
    if std == 0 or not np.isfinite(std):
      return labels_arr
    flat[finite] = (flat[finite] - np.nanmean(flat[finite])) / std
    return flat[:, None]


class NormalizeLabels(OutputWarper):

  def __init__(self, target_interval: Tuple[float, float] = (0.0, 1.0)):
    if target_interval[0] > target_interval[1]:
      raise ValueError(f'Bounds {target_interval} invalid.')
    self.target_interval = target_interval

  def warp(self, labels_arr: np.ndarray) -> np.ndarray:
    labels_arr = _validate(labels_arr)
    flat = labels_arr.flatten()
    if np.all(np.isnan(flat)):
      raise ValueError('Labels need at least one non-NaN entry.')
    finite = np.isfinite(flat)
    lo, hi = np.min(flat[finite]), np.max(flat[finite])
    t0, t1 = self.target_interval
    if lo == hi:
      flat[finite] = 0.5 * (t0 + t1)
    else:
      flat[finite] = t0 + (flat[finite] - lo) * (t1 - t0) / (hi - lo)
    return flat[:, None]


def create_default_warper(*, half_rank_warp: bool = True,
                          log_warp: bool = True,
                          infeasible_warp: bool = True
                          ) -> OutputWarperPipeline:
  if not (half_rank_warp or log_warp or infeasible_warp):
    raise ValueError('At least one warp must be enabled.')
  warpers: List[OutputWarper] = []
  if half_rank_warp:
    warpers.append(HalfRankComponent())
  if log_warp:
    warpers.append(LogWarperComponent())
  if infeasible_warp:
    warpers.append(InfeasibleWarperComponent())
  return OutputWarperPipeline(warpers)
