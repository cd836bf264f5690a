"""Statistical comparison of simple-regret samples.

Capability parity with vizier/_src/benchmarks/analyzers/
simple_regret_score.py:26 (one-sided t-test / score on best-value
samples from repeated runs).
"""

from __future__ import annotations

from typing import Sequence

import numpy as np
from scipy import stats


def t_test_less_mean_score(baseline_samples: Sequence[float],
                           candidate_samples: Sequence[float]) -> float:
  """P-value that candidate mean <= baseline mean (small = candidate wins).

  Samples are 'best objective value found' per run, maximized.
  """
  baseline = np.asarray(baseline_samples, dtype=np.float64)
  candidate = np.asarray(candidate_samples, dtype=np.float64)
  if baseline.std() == 0 and candidate.std() == 0:
    if candidate.mean() > baseline.mean():
      return 0.0
    return 1.0
  result = stats.ttest_ind(candidate, baseline, equal_var=False,
                           alternative='less')
  # 'less' tests candidate < baseline; we want candidate BETTER
  # (greater), so return that p-value directly for the caller to compare.
  return float(result.pvalue)


def t_test_mean_score(baseline_samples: Sequence[float],
                      candidate_samples: Sequence[float]) -> float:
  """P-value of 'candidate mean is NOT greater than baseline mean'.

  Small values mean the candidate is confidently better (maximization).
  """
  baseline = np.asarray(baseline_samples, dtype=np.float64)
  candidate = np.asarray(candidate_samples, dtype=np.float64)
  if baseline.std() == 0 and candidate.std() == 0:
    return 0.0 if candidate.mean() > baseline.mean() else 1.0
  result = stats.ttest_ind(candidate, baseline, equal_var=False,
                           alternative='greater')
  return float(result.pvalue)
