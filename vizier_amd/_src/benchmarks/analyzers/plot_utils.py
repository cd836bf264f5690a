"""Matplotlib helpers for benchmark curves and records.

Capability parity with vizier/_src/benchmarks/analyzers/plot_utils.py
(plot_median_convergence :31, plot_mean_convergence :80,
plot_from_records :125).
"""

from __future__ import annotations

from typing import Optional, Sequence

import numpy as np

from vizier_amd._src.benchmarks.analyzers.convergence_curve import (
    ConvergenceCurve,
)
from vizier_amd._src.benchmarks.analyzers.state_analyzer import (
    BenchmarkRecord,
)


def plot_median_convergence(ax, curve: ConvergenceCurve, *,
                            percentiles=(25, 75), label: Optional[str] =
                            None, color=None):
  """Median line with a percentile band."""
  med = np.nanmedian(curve.ys, axis=0)
  lo = np.nanpercentile(curve.ys, percentiles[0], axis=0)
  hi = np.nanpercentile(curve.ys, percentiles[1], axis=0)
  line, = ax.plot(curve.xs, med, label=label, color=color)
  ax.fill_between(curve.xs, lo, hi, alpha=0.2, color=line.get_color())
  ax.set_xlabel('Num Trials')
  if curve.ylabel:
    ax.set_ylabel(curve.ylabel)
  return ax


def plot_mean_convergence(ax, curve: ConvergenceCurve, *,
                          label: Optional[str] = None, color=None):
  """Mean line with a +-1 stderr band."""
  mean = np.nanmean(curve.ys, axis=0)
  stderr = np.nanstd(curve.ys, axis=0) / max(
      np.sqrt(curve.num_curves), 1.0)
  line, = ax.plot(curve.xs, mean, label=label, color=color)
  ax.fill_between(curve.xs, mean - stderr, mean + stderr, alpha=0.2,
                  color=line.get_color())
  ax.set_xlabel('Num Trials')
  if curve.ylabel:
    ax.set_ylabel(curve.ylabel)
  return ax


def plot_from_records(records: Sequence[BenchmarkRecord], *,
                      metrics: Optional[Sequence[str]] = None,
                      col_figsize: float = 5.0):
  """One column per metric, one line per algorithm; returns (fig, axes)."""
  import matplotlib
  matplotlib.use('Agg')  # headless environments
  import matplotlib.pyplot as plt

  if metrics is None:
    seen = []
    for r in records:
      for name in r.plot_elements:
        if name not in seen:
          seen.append(name)
    metrics = seen
  fig, axes = plt.subplots(
      1, max(len(metrics), 1),
      figsize=(col_figsize * max(len(metrics), 1), 4), squeeze=False)
  for j, metric in enumerate(metrics):
    ax = axes[0][j]
    for rec in records:
      el = rec.plot_elements.get(metric)
      if el is None:
        continue
      curve = getattr(el, 'curve', el)
      plot_type = getattr(el, 'plot_type', 'error-bar')
      if curve is not None and hasattr(curve, 'ys'):
        plot_median_convergence(ax, curve, label=rec.algorithm)
      elif plot_type == 'histogram' and \
          getattr(el, 'plot_array', None) is not None:
        ax.hist(np.asarray(el.plot_array).reshape(-1),
                alpha=0.5, label=rec.algorithm)
    ax.set_title(metric)
    ax.legend()
  return fig, axes
