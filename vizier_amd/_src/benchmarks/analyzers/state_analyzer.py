"""Benchmark records + summaries.

Capability parity with vizier/_src/benchmarks/analyzers/
state_analyzer.py (BenchmarkRecord aggregation) and
exploration_score_utils (entropy-based exploration scores). Plotting is
intentionally out of scope (no matplotlib in the image); records render
to pandas DataFrames instead.
"""

from __future__ import annotations

import collections
import dataclasses
import json
from typing import Dict, List, Optional, Sequence, Tuple

import numpy as np

from vizier_amd import pyvizier as vz
from vizier_amd._src.benchmarks.analyzers.convergence_curve import (
    ConvergenceCurve,
    ConvergenceCurveConverter,
)
from vizier_amd._src.benchmarks.runners.benchmark_state import (
    BenchmarkState,
)


@dataclasses.dataclass
class PlotElement:
  """One subplot's payload (reference state_analyzer.py:37): either a
  curve (error-bar) or a raw array (histogram/scatter)."""

  curve: object = None                 # ConvergenceCurve, for error-bar
  plot_array: object = None            # np.ndarray, for histogram/scatter
  plot_type: str = 'error-bar'         # 'error-bar'|'histogram'|'scatter'
  xlabel: str = 'Num Trials'
  yscale: str = 'linear'
  percentile_error_bar: tuple = (25, 75)


@dataclasses.dataclass
class BenchmarkRecord:
  """One (algorithm, experimenter) benchmark outcome."""

  algorithm: str
  experimenter_metadata: Dict[str, str]
  plot_elements: Dict[str, object] = dataclasses.field(
      default_factory=dict)  # name -> PlotElement or bare curve


class BenchmarkStateAnalyzer:
  """Turns finished BenchmarkStates into records/frames."""

  @staticmethod
  def to_curve(states: Sequence[BenchmarkState],
               flip_signs_for_min: bool = True) -> ConvergenceCurve:
    """Stacks each state's best-so-far curve (one row per repeat)."""
    if not states:
      raise ValueError('No states.')
    curves = []
    for state in states:
      problem = state.experimenter.problem_statement()
      converter = ConvergenceCurveConverter(
          problem.metric_information.item(),
          flip_signs_for_min=flip_signs_for_min)
      trials = state.algorithm.supporter.GetTrials()
      curves.append(converter.convert(trials))
    return ConvergenceCurve.align_xs(curves)

  @staticmethod
  def to_record(algorithm: str, states: Sequence[BenchmarkState]
                ) -> BenchmarkRecord:
    curve = BenchmarkStateAnalyzer.to_curve(states)
    exptr = states[0].experimenter
    return BenchmarkRecord(
        algorithm=algorithm,
        experimenter_metadata={'experimenter': repr(exptr)},
        plot_elements={'objective': curve})

  @staticmethod
  def records_to_frame(records: Sequence[BenchmarkRecord]):
    import pandas as pd
    rows = []
    for r in records:
      final = {}
      for name, el in r.plot_elements.items():
        curve = getattr(el, 'curve', el)
        if curve is not None and hasattr(curve, 'ys'):
          final[name] = float(np.nanmedian(curve.ys[:, -1]))
        elif getattr(el, 'plot_array', None) is not None:
          final[name] = float(np.nanmedian(el.plot_array))
      rows.append({'algorithm': r.algorithm,
                   **r.experimenter_metadata, **final})
    return pd.DataFrame(rows)


def compute_parameter_entropy(trials: Sequence[vz.Trial],
                              config: vz.ParameterConfig,
                              num_bins: Optional[int] = None) -> float:
  """Exploration score: entropy of visited values for one parameter.

  Parity with exploration_score_utils.py:28: categorical/discrete/
  integer parameters count unique values; continuous parameters
  histogram with the cube-root bin rule num_bins = 30/100^(1/3) *
  n^(1/3) (capped at n) unless num_bins is given explicitly.
  """
  values = [t.parameters.get_value(config.name, None) for t in trials]
  values = [v for v in values if v is not None]
  if not values:
    return 0.0
  if config.type in (vz.ParameterType.CATEGORICAL,
                     vz.ParameterType.DISCRETE,
                     vz.ParameterType.INTEGER):
    counts = np.asarray(list(collections.Counter(values).values()),
                        dtype=np.float64)
  else:
    lo, hi = config.bounds
    n = len(values)
    if num_bins is None:
      alpha = 1.0 / 3.0
      num_bins = min(int(30.0 / (100 ** alpha) * n ** alpha), n)
      num_bins = max(num_bins, 1)
    hist, _ = np.histogram([float(v) for v in values], bins=num_bins,
                           range=(lo, hi))
    counts = hist.astype(np.float64)
  p = counts / counts.sum()
  p = p[p > 0]
  return float(-(p * np.log(p)).sum())


def compute_average_marginal_parameter_entropy(
    studies: Sequence[Tuple[vz.ProblemStatement, Sequence[vz.Trial]]]
) -> float:
  """Mean per-parameter entropy across studies
  (exploration_score_utils.py:97)."""
  entropies = []
  for problem, trials in studies:
    for config in problem.search_space.parameters:
      entropies.append(compute_parameter_entropy(trials, config))
  return float(np.mean(entropies)) if entropies else 0.0


class BenchmarkRecordAnalyzer:
  """Cross-record comparisons (reference state_analyzer.py:195)."""

  @classmethod
  def add_comparison_metrics(
      cls, records: Sequence[BenchmarkRecord], baseline_algo: str, *,
      compare_metric: str = 'objective',
      comparator_factory=None) -> List[BenchmarkRecord]:
    """Scores every record's `compare_metric` curve against the
    baseline algorithm's (grouped by experimenter metadata)."""
    from vizier_amd._src.benchmarks.analyzers import convergence_curve as cc
    if comparator_factory is None:
      comparator_factory = \
          cc.LogEfficiencyConvergenceCurveComparatorFactory()

    def curve_of(rec):
      el = rec.plot_elements.get(compare_metric)
      curve = getattr(el, 'curve', el)
      if curve is None:
        raise ValueError(
            f'Record {rec.algorithm} has no curve for {compare_metric}')
      return curve

    def exp_key(rec):
      return json.dumps(dict(rec.experimenter_metadata), sort_keys=True)

    groups: Dict[str, List[BenchmarkRecord]] = {}
    for rec in records:
      groups.setdefault(exp_key(rec), []).append(rec)

    out: List[BenchmarkRecord] = []
    for group in groups.values():
      baselines = [r for r in group if r.algorithm == baseline_algo]
      if not baselines:
        raise ValueError(f'Baseline {baseline_algo} not found in group.')
      base_curve = curve_of(baselines[0])
      for rec in group:
        comparator = comparator_factory(base_curve, curve_of(rec))
        new_elements = dict(rec.plot_elements)
        new_elements[f'{compare_metric}:score'] = PlotElement(
            plot_array=np.asarray([comparator.score()]),
            plot_type='histogram')
        out.append(BenchmarkRecord(
            algorithm=rec.algorithm,
            experimenter_metadata=rec.experimenter_metadata,
            plot_elements=new_elements))
    return out
