"""Statistical designer-vs-baseline comparison harnesses.

Capability parity with vizier/_src/algorithms/testing/comparator_runner.py
(SimpleRegretComparisonTester :54, EfficiencyComparisonTester :120).
"""

from __future__ import annotations

import dataclasses
from typing import Callable

import numpy as np

from vizier_amd import pyvizier as vz
from vizier_amd._src.algorithms.core.abstractions import Designer
from vizier_amd._src.benchmarks.analyzers import simple_regret_score
from vizier_amd._src.benchmarks.experimenters.experimenter import (
    Experimenter,
)
from vizier_amd._src.benchmarks.runners.benchmark_runner import (
    BenchmarkRunner,
    GenerateAndEvaluate,
)
from vizier_amd._src.benchmarks.runners.benchmark_state import (
    BenchmarkState,
)

DesignerFactory = Callable[[vz.ProblemStatement, int], Designer]


class FailedComparisonTestError(Exception):
  """The candidate did not beat the baseline at the required level."""


def _best_values(experimenter: Experimenter, factory: DesignerFactory,
                 *, num_trials: int, num_repeats: int,
                 batch_size: int = 1, seed: int = 0) -> np.ndarray:
  out = []
  for rep in range(num_repeats):
    state = BenchmarkState.from_designer_factory(
        lambda p, _r=rep: factory(p, seed + _r), experimenter)
    BenchmarkRunner([GenerateAndEvaluate(batch_size)],
                    num_repeats=num_trials // batch_size).run(state)
    best = state.algorithm.supporter.GetBestTrials(count=1)
    problem = experimenter.problem_statement()
    mi = problem.metric_information.item()
    value = best[0].final_measurement.metrics[mi.name].value
    out.append(value if mi.goal.is_maximize else -value)
  return np.asarray(out)


@dataclasses.dataclass
class SimpleRegretComparisonTester:
  """Asserts candidate beats baseline on simple regret (t-test)."""

  baseline_num_trials: int
  candidate_num_trials: int
  baseline_suggestion_batch_size: int = 1
  candidate_suggestion_batch_size: int = 1
  baseline_num_repeats: int = 5
  candidate_num_repeats: int = 5
  alpha: float = 0.05

  def assert_benchmark_state_better_simple_regret(
      self, experimenter: Experimenter,
      baseline_factory: DesignerFactory,
      candidate_factory: DesignerFactory) -> None:
    baseline = _best_values(
        experimenter, baseline_factory,
        num_trials=self.baseline_num_trials,
        num_repeats=self.baseline_num_repeats,
        batch_size=self.baseline_suggestion_batch_size)
    candidate = _best_values(
        experimenter, candidate_factory,
        num_trials=self.candidate_num_trials,
        num_repeats=self.candidate_num_repeats,
        batch_size=self.candidate_suggestion_batch_size)
    p = simple_regret_score.t_test_mean_score(baseline, candidate)
    if p > self.alpha:
      raise FailedComparisonTestError(
          f'p-value {p:.4f} > alpha {self.alpha}: candidate '
          f'(mean {candidate.mean():.4f}) is not confidently better than '
          f'baseline (mean {baseline.mean():.4f}).')
