"""Convergence curves + comparators.

Capability parity with vizier/_src/benchmarks/analyzers/
convergence_curve.py (ConvergenceCurve :35, ConvergenceCurveConverter
:255, HypervolumeCurveConverter :342, LogEfficiencyCurveComparator :714,
PercentageBetterComparator :837, WinRateComparator :913).
"""

from __future__ import annotations

import abc
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


class StatefulCurveConverter(abc.ABC):
  """Converter whose state carries across convert() calls, so
  convert(t1 + t2) == convert(t1) ++ convert(t2)
  (reference convergence_curve.py:241)."""

  @abc.abstractmethod
  def convert(self, trials: Sequence[vz.Trial]) -> ConvergenceCurve:
    ...


@dataclasses.dataclass
class ConvergenceCurveConverter(StatefulCurveConverter):
  """Trials -> best-so-far objective curve (stateful across calls)."""

  metric_information: vz.MetricInformation
  flip_signs_for_min: bool = False
  _best: float = dataclasses.field(default=float('nan'), repr=False)
  _cost: int = dataclasses.field(default=0, repr=False)

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
    # Running best, NaN-safe, continuing from previous calls.
    out = np.empty_like(ys)
    best = self._best
    for i, v in enumerate(ys):
      if np.isnan(best):
        best = v
      elif not np.isnan(v):
        best = max(best, v) if maximize else min(best, v)
      out[i] = best
    self._best = best
    xs = np.arange(self._cost + 1, self._cost + len(ys) + 1)
    self._cost += len(ys)
    trend = (ConvergenceCurve.YSign.MAXIMIZE if maximize
             else ConvergenceCurve.YSign.MINIMIZE)
    return ConvergenceCurve(xs, out[None, :], ylabel=name, trend=trend)


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


# -- Reference-style comparator family (convergence_curve.py:560-1100) ----


@dataclasses.dataclass
class ConvergenceComparator(abc.ABC):
  """Holds (baseline, compared) curve pair; subclasses score them.

  Score > 0 means `compared` is better; quantiles select which row of a
  batched curve represents each side.
  """

  baseline_curve: ConvergenceCurve
  compared_curve: ConvergenceCurve
  baseline_quantile: float = 0.5
  compared_quantile: float = 0.5
  name: str = 'score'

  def __post_init__(self):
    if not 0 <= self.baseline_quantile <= 1:
      raise ValueError('baseline_quantile must be in [0, 1]')
    if not 0 <= self.compared_quantile <= 1:
      raise ValueError('compared_quantile must be in [0, 1]')
    if self.baseline_curve.trend != self.compared_curve.trend:
      raise ValueError('Trend mismatch between curves.')

  def _quantile_rows(self):
    b = np.nanquantile(self.baseline_curve.ys, self.baseline_quantile,
                       axis=0)
    c = np.nanquantile(self.compared_curve.ys, self.compared_quantile,
                       axis=0)
    n = min(len(b), len(c))
    return b[:n], c[:n]

  @abc.abstractmethod
  def score(self) -> float:
    ...

  def curve(self) -> ConvergenceCurve:
    raise NotImplementedError


class WinRateConvergenceCurveComparator(ConvergenceComparator):
  """Mean win-rate of `compared` over `baseline`, in [-0.5, 0.5]."""

  def curve(self) -> ConvergenceCurve:
    maximize = (self.baseline_curve.trend ==
                ConvergenceCurve.YSign.MAXIMIZE)
    base = self.baseline_curve.ys
    comp = self.compared_curve.ys
    n = min(base.shape[1], comp.shape[1])
    base, comp = base[:, :n], comp[:, :n]
    if not maximize:
      base, comp = -base, -comp
    wins = np.zeros(n)
    for brow in base:
      wins += (comp > brow).mean(axis=0) + \
          0.5 * (comp == brow).mean(axis=0)
    ys = wins / base.shape[0] - 0.5
    return ConvergenceCurve(self.baseline_curve.xs[:n], ys[None, :],
                            trend=self.baseline_curve.trend)

  def score(self) -> float:
    return float(np.nanmean(self.curve().ys))


class _ScorerComparator(ConvergenceComparator):
  """Adapts the single-argument scorers above to the pair interface."""

  _scorer_cls = None

  def score(self) -> float:
    return float(self._scorer_cls(self.baseline_curve).score(
        self.compared_curve))


class _LogEfficiencyPair(_ScorerComparator):
  _scorer_cls = LogEfficiencyConvergenceCurveComparator


class _PercentageBetterPair(_ScorerComparator):
  _scorer_cls = PercentageBetterConvergenceCurveComparator


class ConvergenceComparatorFactory:
  """Callable protocol: (baseline, compared, ...) -> comparator."""

  _comparator_cls = None

  def __call__(self, baseline_curve: ConvergenceCurve,
               compared_curve: ConvergenceCurve,
               baseline_quantile: float = 0.5,
               compared_quantile: float = 0.5,
               **kwargs) -> ConvergenceComparator:
    return self._comparator_cls(
        baseline_curve=baseline_curve, compared_curve=compared_curve,
        baseline_quantile=baseline_quantile,
        compared_quantile=compared_quantile, **kwargs)


class WinRateConvergenceCurveComparatorFactory(ConvergenceComparatorFactory):
  _comparator_cls = WinRateConvergenceCurveComparator


class LogEfficiencyConvergenceCurveComparatorFactory(
    ConvergenceComparatorFactory):
  _comparator_cls = _LogEfficiencyPair


class PercentageBetterConvergenceCurveComparatorFactory(
    ConvergenceComparatorFactory):
  _comparator_cls = _PercentageBetterPair


@dataclasses.dataclass
class MultiMetricCurveConverter(StatefulCurveConverter):
  """Single- or multi-objective curve converter chosen from a
  MetricsConfig; unsafe trials are warped first (reference :464)."""

  metrics_config: vz.MetricsConfig
  converter: StatefulCurveConverter

  @classmethod
  def from_metrics_config(cls, metrics_config: vz.MetricsConfig,
                          **kwargs) -> 'MultiMetricCurveConverter':
    from vizier_amd._src.pyvizier import multimetric as _mm  # noqa: F401
    objectives = metrics_config.of_type(vz.MetricType.OBJECTIVE)
    if metrics_config.is_single_objective:
      converter = ConvergenceCurveConverter(objectives.item(), **kwargs)
    else:
      converter = HypervolumeCurveConverter(list(objectives), **kwargs)
    return cls(metrics_config, converter)

  def convert(self, trials: Sequence[vz.Trial]) -> ConvergenceCurve:
    if not trials:
      raise ValueError('No trials provided')
    import copy as _copy
    from vizier_amd._src.pyvizier import multimetric as _mm
    checker = _mm.SafetyChecker(self.metrics_config)
    warped = checker.warp_unsafe_trials(_copy.deepcopy(list(trials)))
    return self.converter.convert(warped)


@dataclasses.dataclass
class RestartingCurveConverter(StatefulCurveConverter):
  """Rebuilds the wrapped converter at exponentially-spaced restarts
  so late-study curves reflect all data (reference :516)."""

  converter_factory: 'Callable[[], StatefulCurveConverter]'
  restart_min_trials: int = 10
  restart_rate: float = 2.0
  _all_trials: list = dataclasses.field(default_factory=list)
  _converter: Optional[StatefulCurveConverter] = None

  def convert(self, trials: Sequence[vz.Trial]) -> ConvergenceCurve:
    if self._converter is None:
      self._converter = self.converter_factory()
      if self._all_trials:
        self._converter.convert(self._all_trials)
    curve = self._converter.convert(list(trials))
    self._all_trials.extend(trials)
    if len(self._all_trials) >= self.restart_min_trials:
      import math as _math
      prev = _math.log(1 + len(self._all_trials) - len(trials)) / \
          _math.log(self.restart_rate)
      now = _math.log(1 + len(self._all_trials)) / \
          _math.log(self.restart_rate)
      if int(now) > int(prev):
        self._converter = None
    return curve
