"""Trial <-> feature-array converters feeding the GP/Eagle compute path.

Capability parity with vizier/pyvizier/converters/core.py
(DefaultModelInputConverter :539, DefaultModelOutputConverter :788,
TrialToArrayConverter :1217) and jnp_converters.py:147
(TrialToModelInputConverter: discrete/int parameters are *continuified*
into [0,1] and rounded back to feasible points on `to_parameters`;
categoricals are one-hot embedded), plus padding.py:28-97
(PaddingSchedule) to keep kernel shapes stable for the HIP path.

The output is a single dense float matrix in [0,1]^(N x D): the native
layout the MI355X Gram/predict kernels consume directly.
"""

from __future__ import annotations

import dataclasses
import enum
import math
from typing import Any, Dict, List, Optional, Sequence, Tuple, Union

import numpy as np

from vizier_amd import pyvizier as vz


class PaddingType(enum.Enum):
  NONE = 'NONE'
  MULTIPLES_OF_10 = 'MULTIPLES_OF_10'
  POWERS_OF_2 = 'POWERS_OF_2'


@dataclasses.dataclass(frozen=True)
class PaddingSchedule:
  """Pads (trials, features, metrics) so kernel shapes re-use.

  Parity with vizier/pyvizier/converters/padding.py:28-97: each axis
  buckets independently (NONE / MULTIPLES_OF_10 / POWERS_OF_2). Trials
  bucketing stabilizes shapes across suggest calls (Gram/Cholesky/
  hipGraph re-use); feature bucketing stabilizes D across studies in
  one process (conditional spaces, transfer stacks) — zero-padded
  feature columns are shared by every row, so they contribute 0 to all
  pairwise distances and the GP math is unchanged; metric bucketing
  pads labels with NaN (masked as missing).
  """

  num_trials: PaddingType = PaddingType.NONE
  num_features: PaddingType = PaddingType.NONE
  num_metrics: PaddingType = PaddingType.NONE

  @staticmethod
  def _bucket(n: int, kind: PaddingType) -> int:
    if n == 0 or kind == PaddingType.NONE:
      return n
    if kind == PaddingType.MULTIPLES_OF_10:
      return int(math.ceil(n / 10.0) * 10)
    return 1 << (n - 1).bit_length()

  def padded_size(self, n: int) -> int:
    """Trials-axis bucket (back-compat name)."""
    return self._bucket(n, self.num_trials)

  def padded_features(self, d: int) -> int:
    return self._bucket(d, self.num_features)

  def padded_metrics(self, m: int) -> int:
    return self._bucket(m, self.num_metrics)


def pad_rows(arr: np.ndarray, target: int,
             fill: float = 0.0) -> Tuple[np.ndarray, np.ndarray]:
  """Pads axis 0 to `target`; returns (padded, is_valid_mask)."""
  n = arr.shape[0]
  mask = np.zeros(target, dtype=bool)
  mask[:n] = True
  if target == n:
    return arr, mask
  pad_shape = (target - n,) + arr.shape[1:]
  return np.concatenate([arr, np.full(pad_shape, fill, arr.dtype)]), mask


@dataclasses.dataclass
class _ColumnSpec:
  """How one parameter maps into feature columns."""

  config: vz.ParameterConfig
  start: int
  width: int  # 1 for numeric, num categories for one-hot.

  @property
  def is_onehot(self) -> bool:
    return self.width > 1 or \
        self.config.type == vz.ParameterType.CATEGORICAL


def _scale(config: vz.ParameterConfig, value: float) -> float:
  lo, hi = config.bounds
  if hi == lo:
    return 0.0
  st = config.scale_type
  if st == vz.ScaleType.LOG and lo > 0:
    return (math.log(value) - math.log(lo)) / (math.log(hi) - math.log(lo))
  if st == vz.ScaleType.REVERSE_LOG and lo > 0:
    flipped = hi + lo - value
    return 1.0 - (math.log(flipped) - math.log(lo)) / (math.log(hi) -
                                                       math.log(lo))
  return (value - lo) / (hi - lo)


def _unscale(config: vz.ParameterConfig, u: float) -> float:
  lo, hi = config.bounds
  u = min(max(u, 0.0), 1.0)
  if hi == lo:
    return lo
  st = config.scale_type
  if st == vz.ScaleType.LOG and lo > 0:
    return math.exp(math.log(lo) + u * (math.log(hi) - math.log(lo)))
  if st == vz.ScaleType.REVERSE_LOG and lo > 0:
    flipped = math.exp(math.log(lo) + (1.0 - u) * (math.log(hi) -
                                                   math.log(lo)))
    return hi + lo - flipped
  return lo + u * (hi - lo)


class TrialToArrayConverter:
  """Flattens trials into a dense [0,1] feature matrix and a label matrix.

  Labels are sign-flipped so that 'larger is better' for every metric
  (DefaultModelOutputConverter behavior). Conditional children are
  flattened; values missing from a trial are imputed at 0.5 (numeric) or
  all-zero one-hot.
  """

  def __init__(self, problem: vz.ProblemStatement, *,
               pad_oovs: bool = False,
               padding_schedule: Optional[PaddingSchedule] = None,
               dtype=np.float32):
    self._problem = problem
    self._padding = padding_schedule or PaddingSchedule()
    self._dtype = dtype
    self._columns: List[_ColumnSpec] = []
    offset = 0
    configs: List[vz.ParameterConfig] = []
    for top in problem.search_space.parameters:
      configs.extend(top.traverse())
    for cfg in configs:
      if cfg.type == vz.ParameterType.CATEGORICAL:
        width = len(cfg.feasible_values)
      else:
        width = 1
      self._columns.append(_ColumnSpec(cfg, offset, width))
      offset += width
    self._n_features = offset
    self._metrics = list(problem.metric_information)

  @property
  def n_features(self) -> int:
    return self._n_features

  @property
  def n_labels(self) -> int:
    return len(self._metrics)

  @property
  def metric_information(self) -> List[vz.MetricInformation]:
    return list(self._metrics)

  @property
  def output_specs(self) -> List[_ColumnSpec]:
    return list(self._columns)

  # -- forward --------------------------------------------------------------

  def to_features(self, trials: Sequence[vz.TrialSuggestion]) -> np.ndarray:
    out = np.zeros((len(trials), self._n_features), dtype=self._dtype)
    for i, trial in enumerate(trials):
      for col in self._columns:
        value = trial.parameters.get_value(col.config.name, None)
        if col.is_onehot:
          if value is not None:
            try:
              idx = col.config.feasible_values.index(value)
              out[i, col.start + idx] = 1.0
            except ValueError:
              pass  # Out-of-vocabulary: all-zero row.
        else:
          if value is None:
            out[i, col.start] = 0.5
          else:
            out[i, col.start] = _scale(col.config, float(value))
    return out

  def to_labels(self, trials: Sequence[vz.Trial]) -> np.ndarray:
    """(N, M) matrix; NaN for missing metrics; flipped to maximize."""
    out = np.full((len(trials), len(self._metrics)), np.nan,
                  dtype=self._dtype)
    for i, trial in enumerate(trials):
      if trial.final_measurement is None:
        continue
      for j, mi in enumerate(self._metrics):
        metric = trial.final_measurement.metrics.get(mi.name)
        if metric is None:
          continue
        out[i, j] = metric.value if mi.goal.is_maximize else -metric.value
    return out

  def to_xy(self, trials: Sequence[vz.Trial]
            ) -> Tuple[np.ndarray, np.ndarray]:
    return self.to_features(trials), self.to_labels(trials)

  def to_padded_xy(self, trials: Sequence[vz.Trial]
                   ) -> Tuple[np.ndarray, np.ndarray, np.ndarray]:
    """Returns (features, labels, valid_mask) padded per the schedule.

    All three axes bucket: trials (rows, mask returned), features
    (zero columns — distance-neutral for the GP kernels), metrics
    (NaN columns — masked as missing labels).
    """
    x, y = self.to_xy(trials)
    target = self._padding.padded_size(len(trials))
    x, mask = pad_rows(x, target)
    y, _ = pad_rows(y, target, fill=np.nan)
    df = self._padding.padded_features(x.shape[1])
    if df > x.shape[1]:
      x = np.concatenate(
          [x, np.zeros((x.shape[0], df - x.shape[1]), x.dtype)], axis=1)
    dm = self._padding.padded_metrics(y.shape[1])
    if dm > y.shape[1]:
      y = np.concatenate(
          [y, np.full((y.shape[0], dm - y.shape[1]), np.nan, y.dtype)],
          axis=1)
    return x, y, mask

  # -- inverse --------------------------------------------------------------

  def to_parameters(self, array: np.ndarray) -> List[vz.ParameterDict]:
    array = np.asarray(array, dtype=np.float64)
    if array.ndim == 1:
      array = array[None, :]
    out = []
    for row in array:
      params = vz.ParameterDict()
      for col in self._columns:
        cfg = col.config
        if col.is_onehot:
          idx = int(np.argmax(row[col.start:col.start + col.width]))
          params[cfg.name] = cfg.feasible_values[idx]
        else:
          value = _unscale(cfg, float(row[col.start]))
          if cfg.type in (vz.ParameterType.INTEGER,
                          vz.ParameterType.DISCRETE):
            value = cfg.round_to_feasible(value)
          params[cfg.name] = value
      out.append(params)
    return out

  def to_suggestions(self, array: np.ndarray) -> List[vz.TrialSuggestion]:
    return [vz.TrialSuggestion(p) for p in self.to_parameters(array)]
