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
    # Match the reference (output_warpers.py InfeasibleWarperComponent):
    # substitute warped_bad for NaNs FIRST, then shift ALL entries so that
    # E[warp] = 0 holds over the substituted array (and unwarp() below,
    # which subtracts the shift from everything, is the exact inverse).
    nan_mask = np.isnan(flat)
    flat[nan_mask] = warped_bad
    flat += self._shift
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


class DetectOutliers(OutputWarper):
  """Marks unreasonably-bad finite labels as NaN (infeasible).

  Parity with output_warpers.py:578 (DetectOutliers). The variance of
  the good half is estimated from (max - median) with the sample-size
  dependent divisors of Hozo et al. 2005 (the reference's source);
  labels below median - min_zscore * std become NaN so the infeasible
  warper can handle them.
  """

  def __init__(self, *, min_zscore: float = 6.0,
               max_zscore: Optional[float] = None):
    self.min_zscore = min_zscore
    self.max_zscore = max_zscore

  def _estimate_variance(self, labels_arr: np.ndarray) -> float:
    num_points = labels_arr.shape[0]
    labels_median = float(np.nanmedian(labels_arr))
    labels_max = float(np.nanmax(labels_arr))
    if not np.isfinite(labels_max):
      raise ValueError('The max label value should be finite.')
    if not np.isfinite(labels_median):
      raise ValueError('The median label value should be finite.')
    if self.max_zscore:
      return (labels_max - labels_median) / self.min_zscore
    if num_points >= 70:
      return (labels_max - labels_median) / 3
    if num_points >= 15:
      return (labels_max - labels_median) / 2
    # Small-sample range-based estimator (eq. 12 of the paper above),
    # hallucinating the min at median - max as the reference does.
    labels_min = labels_median - np.max(labels_arr)
    if labels_min < 0:
      labels_min = 0.0
    a, m, b, n = labels_min, labels_median, labels_max, num_points
    out = a ** 2 + m ** 2 + b ** 2
    out += ((n - 3) / 2) * ((a + m) ** 2 + (b + m) ** 2) / 4
    out -= n * ((a + 2 * m + b) / 4 + (a - 2 * m + b) / (4 * n)) ** 2
    return out / (n - 1)

  def warp(self, labels_arr: np.ndarray) -> np.ndarray:
    labels_arr = _validate(labels_arr)
    finite = np.isfinite(labels_arr)
    vals = labels_arr[finite]
    median = np.median(vals)
    std = np.sqrt(self._estimate_variance(vals))
    threshold = median - self.min_zscore * std
    vals[vals < threshold] = np.nan
    labels_arr[finite] = vals
    return labels_arr


def _softclip(x: np.ndarray, low: float, high: float,
              softness: float) -> np.ndarray:
  """Smooth clip of x into (low, high): softplus-based, identity for
  interior values when softness is small (TFP SoftClip semantics)."""
  sp = lambda t: np.logaddexp(0.0, t)  # softplus, overflow-safe
  return low + softness * sp((x - low) / softness) \
      - softness * sp((x - high) / softness)


class TransformToGaussian(OutputWarper):
  """Quantile-transforms labels toward a standard Gaussian.

  Parity with output_warpers.py:666. Labels (or their ranks when
  use_rank) are min-max normalized, soft-clipped into (0, 1), and
  mapped through the normal quantile function.
  """

  def __init__(self, *, softclip_low: float = 1e-10,
               softclip_high: float = 1 - 1e-10,
               softclip_hinge_softness: float = 0.01,
               use_rank: bool = False):
    self.softclip_low = softclip_low
    self.softclip_high = softclip_high
    self.softclip_hinge_softness = softclip_hinge_softness
    self.use_rank = use_rank

  def warp(self, labels_arr: np.ndarray) -> np.ndarray:
    labels_arr = _validate(labels_arr)
    flat = labels_arr.flatten()
    base = np.argsort(flat).astype(np.float64) if self.use_rank else flat
    span = np.max(base) - np.min(base)
    normalized = (base - np.min(base)) / span if span > 0 \
        else np.zeros_like(base)
    clipped = _softclip(normalized, self.softclip_low,
                        self.softclip_high,
                        self.softclip_hinge_softness)
    from scipy import special
    return special.ndtri(clipped).reshape(labels_arr.shape)


class LinearOutputWarper:
  """Invertible affine map of each metric column to [low, high].

  Parity with output_warpers.py:729 (equinox LinearOutputWarper):
  per-metric min/max from observations, slope floored by min_range.
  Works on numpy arrays and torch tensors alike (pure arithmetic).
  """

  def __init__(self, *, low_bound, high_bound, min_value, max_value,
               min_range):
    self.low_bound = low_bound
    self.high_bound = high_bound
    self.min_value = min_value
    self.max_value = max_value
    self.min_range = min_range

  @classmethod
  def from_obs(cls, y_obs, low_bound: float = -2.0,
               high_bound: float = 2.0,
               min_range: float = 1e-20) -> 'LinearOutputWarper':
    min_value = y_obs.min(axis=0) if isinstance(y_obs, np.ndarray) \
        else y_obs.min(dim=0).values
    max_value = y_obs.max(axis=0) if isinstance(y_obs, np.ndarray) \
        else y_obs.max(dim=0).values
    return cls(low_bound=low_bound, high_bound=high_bound,
               min_value=min_value, max_value=max_value,
               min_range=min_range)

  def _slope(self):
    rng = self.max_value - self.min_value
    if isinstance(rng, np.ndarray):
      rng = np.maximum(rng, self.min_range)
    else:
      rng = rng.clamp_min(self.min_range)
    return (self.high_bound - self.low_bound) / rng

  def warp(self, y):
    return (y - self.min_value) * self._slope() + self.low_bound

  def unwarp(self, y):
    return (y - self.low_bound) / self._slope() + self.min_value


def create_warp_outliers_warper(
    *, warp_outliers: bool = True, infeasible_warp: bool = True,
    transform_gaussian: bool = True) -> OutputWarperPipeline:
  """Outlier-robust pipeline (output_warpers.py:216)."""
  warpers: List[OutputWarper] = []
  if warp_outliers:
    warpers.append(DetectOutliers())
  if infeasible_warp:
    warpers.append(InfeasibleWarperComponent())
  if transform_gaussian:
    warpers.append(TransformToGaussian())
  return OutputWarperPipeline(warpers)
