"""k-of-n convergence gate on the simplekd testbed.

Capability parity with vizier/_src/algorithms/testing/simplekd_runner.py
(SimpleKDConvergenceTester :68-108).
"""

from __future__ import annotations

import dataclasses
from typing import Callable

from vizier_amd import pyvizier as vz
from vizier_amd._src.algorithms.core.abstractions import Designer
from vizier_amd._src.benchmarks.experimenters.synthetic.simplekd import (
    SimpleKDExperimenter,
)
from vizier_amd._src.benchmarks.runners.benchmark_runner import (
    BenchmarkRunner,
    GenerateAndEvaluate,
)
from vizier_amd._src.benchmarks.runners.benchmark_state import (
    BenchmarkState,
)


class ConvergenceFailure(Exception):
  pass


@dataclasses.dataclass
class SimpleKDConvergenceTester:
  """Requires >= num_required of num_seeds runs to converge."""

  best_category: str
  designer_factory: Callable[[vz.ProblemStatement, int], Designer]
  num_trials: int
  max_relative_error: float = 0.05
  num_seeds: int = 3
  num_required: int = 2
  batch_size: int = 1

  def assert_convergence(self) -> None:
    successes = 0
    details = []
    for seed in range(self.num_seeds):
      experimenter = SimpleKDExperimenter(self.best_category)
      state = BenchmarkState.from_designer_factory(
          lambda p, _s=seed: self.designer_factory(p, _s), experimenter)
      BenchmarkRunner(
          [GenerateAndEvaluate(self.batch_size)],
          num_repeats=self.num_trials // self.batch_size).run(state)
      best = state.algorithm.supporter.GetBestTrials(count=1)[0]
      value = best.final_measurement.metrics['value'].value
      optimum = experimenter.optimal_value
      rel_err = abs(optimum - value) / abs(optimum)
      ok = (rel_err <= self.max_relative_error and
            best.parameters.get_value('categorical') ==
            self.best_category)
      successes += ok
      details.append(f'seed {seed}: best={value:.3f} rel_err={rel_err:.3f}')
    if successes < self.num_required:
      raise ConvergenceFailure(
          f'Only {successes}/{self.num_seeds} runs converged '
          f'(needed {self.num_required}): {details}')
