"""Spatio-temporal converters: trials with intermediate measurements.

Capability parity with vizier/pyvizier/converters/spatio_temporal.py:
- TimedLabels / TimedLabelsExtractor (:31,:43): per-trial time series of
  metric values with 'raw' / 'cummax' / 'cummax_lastonly' /
  'cummax_firstonly' extraction and 'steps' / 'elapsed_secs' / 'index'
  timestamps, optionally resampled at fixed temporal index points.
- SparseSpatioTemporalConverter (:234): one (features + timestamp) row
  per measurement — for unaligned time grids.
- DenseSpatioTemporalConverter (:341): aligned (n_trials, T) label
  matrix over a shared temporal grid, NaN where a trial has no
  observation yet — for batched GPU regression over curves.

Built on the dense TrialToArrayConverter (vizier_amd/converters/core)
rather than the reference's per-parameter dict layout: a single
(N, n_features[+1]) matrix feeds torch directly.
"""

from __future__ import annotations

import dataclasses
from typing import Dict, List, Literal, Optional, Sequence, Tuple

import numpy as np

from vizier_amd import pyvizier as vz
from vizier_amd.converters.core import TrialToArrayConverter

CUMMAX = 'cummax'
CUMMAX_LASTONLY = 'cummax_lastonly'
CUMMAX_FIRSTONLY = 'cummax_firstonly'
RAW = 'raw'


@dataclasses.dataclass
class TimedLabels:
  """One trial's (M, 1) timestamps and per-metric (M, 1) values."""

  times: np.ndarray
  labels: Dict[str, np.ndarray]


class TimedLabelsExtractor:
  """Extracts per-trial metric time series (spatio_temporal.py:43)."""

  def __init__(self, metrics: Sequence[vz.MetricInformation],
               timestamp: Literal['steps', 'elapsed_secs', 'index'] =
               'steps', *,
               temporal_index_points: Sequence[float] = (),
               value_extraction: str = CUMMAX_LASTONLY):
    if value_extraction not in (RAW, CUMMAX, CUMMAX_LASTONLY,
                                CUMMAX_FIRSTONLY):
      raise ValueError(f'Bad value_extraction: {value_extraction}')
    if timestamp not in ('steps', 'elapsed_secs', 'index'):
      raise ValueError(f'Invalid timestamp: {timestamp}')
    self.metrics = list(metrics)
    self.timestamp = timestamp
    self.temporal_index_points = np.asarray(temporal_index_points,
                                            dtype=float).reshape(-1)
    self.value_extraction = value_extraction
    if value_extraction in (CUMMAX_LASTONLY, CUMMAX_FIRSTONLY):
      if len(self.metrics) > 1:
        raise ValueError(f'{value_extraction} supports a single metric.')
      if self.temporal_index_points.size:
        raise ValueError(
            f'{value_extraction} does not support fixed index points.')

  def _acc_fn(self, metric: vz.MetricInformation):
    return (np.maximum if metric.goal == vz.ObjectiveMetricGoal.MAXIMIZE
            else np.minimum)

  def _improved(self, metric: vz.MetricInformation, arr: np.ndarray
                ) -> np.ndarray:
    """arr[i+1] strictly better than arr[i] under the metric's goal."""
    if metric.goal == vz.ObjectiveMetricGoal.MAXIMIZE:
      return arr[:-1] < arr[1:]
    return arr[:-1] > arr[1:]

  def to_timestamps(self, measurements: Sequence[vz.Measurement]
                    ) -> np.ndarray:
    if self.timestamp == 'steps':
      ts = [m.steps for m in measurements]
    elif self.timestamp == 'elapsed_secs':
      ts = [m.elapsed_secs for m in measurements]
    else:
      ts = list(range(len(measurements)))
    return np.asarray(ts, dtype=float)[:, None]

  def extract_all_timestamps(self, trials: Sequence[vz.Trial]
                             ) -> List[float]:
    out: set = set()
    for t in trials:
      out.update(self.to_timestamps(t.measurements).flatten())
    return sorted(out)

  def _metric_values(self, metric: vz.MetricInformation,
                     measurements: Sequence[vz.Measurement]) -> np.ndarray:
    vals = [m.metrics[metric.name].value if metric.name in m.metrics
            else np.nan for m in measurements]
    return np.asarray(vals, dtype=float)[:, None]

  def convert(self, trials: Sequence[vz.Trial]) -> List[TimedLabels]:
    out = []
    for trial in trials:
      times = self.to_timestamps(trial.measurements)
      labels: Dict[str, np.ndarray] = {}
      for metric in self.metrics:
        raw = self._metric_values(metric, trial.measurements)
        if self.value_extraction == RAW:
          vals = raw
        else:
          vals = self._acc_fn(metric).accumulate(raw, axis=0)
        if self.value_extraction in (CUMMAX_LASTONLY, CUMMAX_FIRSTONLY):
          flat = vals.reshape(-1)
          if flat.size:
            if self.value_extraction == CUMMAX_LASTONLY:
              # Record the measurement just before each improvement,
              # plus the final one.
              keep = np.concatenate(
                  [self._improved(metric, flat), [True]])
            else:
              # Record each improvement itself, plus the final one.
              keep = np.concatenate(
                  [[True], self._improved(metric, flat)])
              keep[-1] = True
          else:
            keep = np.zeros(0, dtype=bool)
          labels[metric.name] = flat[keep][:, None]
          times = times[keep]
        else:
          labels[metric.name] = vals
      if (self.temporal_index_points.size and
          self.value_extraction == RAW):
        mask = np.isin(times.reshape(-1), self.temporal_index_points)
        times = times[mask]
        labels = {k: v[mask] for k, v in labels.items()}
      elif (self.temporal_index_points.size and
            self.value_extraction == CUMMAX):
        # Last observation at or before each index point (ffill).
        flat_t = self.to_timestamps(trial.measurements).reshape(-1)
        idx = np.searchsorted(flat_t, self.temporal_index_points,
                              side='right') - 1
        idx = np.clip(idx, 0, max(len(flat_t) - 1, 0))
        labels = {k: v[idx] for k, v in labels.items()}
        times = self.temporal_index_points[:, None]
      out.append(TimedLabels(times, labels))
    return out


class SparseSpatioTemporalConverter:
  """One row per measurement: [trial features, timestamp] -> label.

  For unaligned time grids (spatio_temporal.py:234). The timestamp is
  an extra trailing feature column.
  """

  def __init__(self, converter: TrialToArrayConverter,
               extractor: TimedLabelsExtractor):
    self.converter = converter
    self.extractor = extractor

  @property
  def n_features(self) -> int:
    return self.converter.n_features + 1

  def to_xy(self, trials: Sequence[vz.Trial]
            ) -> Tuple[np.ndarray, np.ndarray]:
    """(sum_i M_i, n_features+1) features and (sum_i M_i, n_metrics)."""
    timed = self.extractor.convert(trials)
    xs, ys = [], []
    for trial, tl in zip(trials, timed):
      m = tl.times.shape[0]
      if m == 0:
        continue
      base = self.converter.to_features([trial])     # (1, F)
      xs.append(np.concatenate(
          [np.tile(base, (m, 1)), tl.times], axis=1))
      ys.append(np.concatenate(
          [tl.labels[mi.name] for mi in self.extractor.metrics], axis=1))
    if not xs:
      return (np.zeros((0, self.n_features)),
              np.zeros((0, len(self.extractor.metrics))))
    return np.concatenate(xs, axis=0), np.concatenate(ys, axis=0)

  def to_features(self, trial: vz.TrialSuggestion,
                  temporal_index_points: np.ndarray) -> np.ndarray:
    """A single trial replicated at the given index points."""
    pts = np.asarray(temporal_index_points, dtype=float).reshape(-1, 1)
    base = self.converter.to_features([trial])
    return np.concatenate([np.tile(base, (pts.shape[0], 1)), pts], axis=1)


class DenseSpatioTemporalConverter:
  """Aligned (n_trials, T) label matrix over a shared temporal grid.

  NaN marks grid points a trial has not reported (yet). Single-metric
  (like the reference's temporal observation matrix,
  spatio_temporal.py:341-475); the matrix feeds batched torch curve
  models directly.
  """

  def __init__(self, converter: TrialToArrayConverter,
               extractor: TimedLabelsExtractor,
               temporal_index_points: Optional[Sequence[float]] = None):
    if len(extractor.metrics) != 1:
      raise ValueError('DenseSpatioTemporalConverter is single-metric.')
    self.converter = converter
    self.extractor = extractor
    self.temporal_index_points = (
        None if temporal_index_points is None
        else np.asarray(temporal_index_points, dtype=float).reshape(-1))

  def to_xty(self, trials: Sequence[vz.Trial]
             ) -> Tuple[np.ndarray, np.ndarray, np.ndarray]:
    """Returns (x: (N, F), t: (T,), y: (N, T) with NaN holes)."""
    grid = (self.temporal_index_points
            if self.temporal_index_points is not None
            else np.asarray(self.extractor.extract_all_timestamps(trials)))
    x = self.converter.to_features(trials)
    y = np.full((len(trials), grid.size), np.nan)
    timed = self.extractor.convert(trials)
    name = self.extractor.metrics[0].name
    for i, tl in enumerate(timed):
      ts = tl.times.reshape(-1)
      pos = np.searchsorted(grid, ts)
      ok = (pos < grid.size)
      ok[ok] &= np.isclose(grid[pos[ok]], ts[ok])
      y[i, pos[ok]] = tl.labels[name].reshape(-1)[ok]
    return x, grid, y

  def to_xy(self, trials: Sequence[vz.Trial]
            ) -> Tuple[np.ndarray, np.ndarray]:
    x, _, y = self.to_xty(trials)
    return x, y
