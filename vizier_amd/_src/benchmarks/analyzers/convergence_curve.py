"""Convergence curves + comparators.

Capability parity with vizier/_src/benchmarks/analyzers/
convergence_curve.py (ConvergenceCurve :35, ConvergenceCurveConverter
:255, HypervolumeCurveConverter :342, LogEfficiencyCurveComparator :714,
PercentageBetterComparator :837, WinRateComparator :913).
"""

from __future__ import annotations

import dataclasses
import enum
from typing import List, Optional, Sequence

import numpy as np

from vizier_amd import pyvizier as vz
from vizier_amd._src.pyvizier import multimetric


class ConvergenceCurve:
  """Best-so-far trajectories: ys (num_curves, num_steps)."""

  class YSign(enum.Enum):
    MAXIMIZE = 'max'
    MINIMIZE = 'min'

  def __init__(self, xs: np.ndarray, ys: np.ndarray,
               ylabel: str = '', trend: 'ConvergenceCurve.YSign' = None):
    self.xs = np.asarray(xs, dtype=np.float64)
    self.ys = np.asarray(ys, dtype=np.float64)
    if self.ys.ndim == 1:
      self.ys = self.ys[None, :]
    if self.xs.shape[0] != self.ys.shape[1]:
      raise ValueError(f'xs {self.xs.shape} vs ys {self.ys.shape}')
    self.ylabel = ylabel
    self.trend = trend or ConvergenceCurve.YSign.MAXIMIZE

  @property
  def num_curves(self) -> int:
    return self.ys.shape[0]

  @classmethod
  def align_xs(cls, curves: Sequence['ConvergenceCurve'],
               *, interpolate_repeats: bool = False) -> 'ConvergenceCurve':
    """Stacks curves onto a common x grid (pads with the last value)."""
    del interpolate_repeats
    if not curves:
      raise ValueError('No curves.')
    max_len = max(len(c.xs) for c in curves)
    xs = np.arange(1, max_len + 1, dtype=np.float64)
    rows = []
    for c in curves:
      for row in c.ys:
        padded = np.concatenate(
            [row, np.full(max_len - len(row), row[-1] if len(row)
                          else np.nan)])
        rows.append(padded)
    return cls(xs, np.stack(rows), ylabel=curves[0].ylabel,
               trend=curves[0].trend)


@dataclasses.dataclass
class ConvergenceCurveConverter:
  """Trials -> best-so-far objective curve."""

  metric_information: vz.MetricInformation
  flip_signs_for_min: bool = False

  def convert(self, trials: Sequence[vz.Trial]) -> ConvergenceCurve:
    name = self.metric_information.name
    maximize = self.metric_information.goal.is_maximize
    values = []
    for t in trials:
      if t.final_measurement is None or name not in \
          t.final_measurement.metrics:
        values.append(np.nan)
      else:
        values.append(t.final_measurement.metrics[name].value)
    ys = np.asarray(values, dtype=np.float64)
    if not maximize and self.flip_signs_for_min:
      ys = -ys
      maximize = True
    # Running best, NaN-safe.
    out = np.empty_like(ys)
    best = np.nan
    for i, v in enumerate(ys):
      if np.isnan(best):
        best = v
      elif not np.isnan(v):
        best = max(best, v) if maximize else min(best, v)
      out[i] = best
    trend = (ConvergenceCurve.YSign.MAXIMIZE if maximize
             else ConvergenceCurve.YSign.MINIMIZE)
    return ConvergenceCurve(np.arange(1, len(ys) + 1), out[None, :],
                            ylabel=name, trend=trend)


@dataclasses.dataclass
class HypervolumeCurveConverter:
  """Trials -> cumulative dominated-hypervolume curve."""

  metrics_information: Sequence[vz.MetricInformation]
  reference_value: Optional[np.ndarray] = None
  num_vectors: int = 1000
  seed: int = 0

  def convert(self, trials: Sequence[vz.Trial]) -> ConvergenceCurve:
    rows = []
    for t in trials:
      if t.final_measurement is None:
        rows.append([np.nan] * len(self.metrics_information))
        continue
      row = []
      for mi in self.metrics_information:
        m = t.final_measurement.metrics.get(mi.name)
        v = m.value if m is not None else np.nan
        row.append(v if mi.goal.is_maximize else -v)
      rows.append(row)
    ys = np.asarray(rows, dtype=np.float64)
    valid = ~np.isnan(ys).any(axis=1)
    origin = (self.reference_value if self.reference_value is not None
              else np.nanmin(np.where(valid[:, None], ys, np.nan),
                             axis=0))
    points = np.where(valid[:, None], ys, origin)
    front = multimetric.ParetoFrontier(points, origin,
                                       num_vectors=self.num_vectors,
                                       seed=self.seed)
    cum = front.hypervolume(is_cumulative=True)
    return ConvergenceCurve(np.arange(1, len(trials) + 1), cum[None, :],
                            ylabel='hypervolume')


def _median_curve(curve: ConvergenceCurve) -> np.ndarray:
  return np.nanmedian(curve.ys, axis=0)


@dataclasses.dataclass
class LogEfficiencyConvergenceCurveComparator:
  """Log of the sample-efficiency ratio vs a baseline curve.

  A positive score means the `compared` curve reaches the baseline's
  final value with fewer trials (log ratio of trial counts).
  """

  baseline_curve: ConvergenceCurve

  def score(self, compared: ConvergenceCurve) -> float:
    base = _median_curve(self.baseline_curve)
    comp = _median_curve(compared)
    maximize = self.baseline_curve.trend == ConvergenceCurve.YSign.MAXIMIZE
    target = base[-1]
    ok = comp >= target if maximize else comp <= target
    if not ok.any():
      return -float(np.log(len(base)))
    t_comp = int(np.argmax(ok)) + 1
    return float(np.log(len(base)) - np.log(t_comp))


@dataclasses.dataclass
class PercentageBetterConvergenceCurveComparator:
  """Fraction of steps where `compared` beats the baseline median."""

  baseline_curve: ConvergenceCurve

  def score(self, compared: ConvergenceCurve) -> float:
    base = _median_curve(self.baseline_curve)
    comp = _median_curve(compared)
    n = min(len(base), len(comp))
    maximize = self.baseline_curve.trend == ConvergenceCurve.YSign.MAXIMIZE
    better = comp[:n] > base[:n] if maximize else comp[:n] < base[:n]
    return float(np.mean(better))


@dataclasses.dataclass
class WinRateComparator:
  """Pairwise win rate between two curve batches at the final step."""

  baseline_curve: ConvergenceCurve

  def score(self, compared: ConvergenceCurve) -> float:
    maximize = self.baseline_curve.trend == ConvergenceCurve.YSign.MAXIMIZE
    base_final = self.baseline_curve.ys[:, -1]
    comp_final = compared.ys[:, -1]
    wins = 0
    total = 0
    for b in base_final:
      for c in comp_final:
        total += 1
        wins += (c > b) if maximize else (c < b)
    return float(wins / max(total, 1))
