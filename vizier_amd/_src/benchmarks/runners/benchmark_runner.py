"""Composable benchmark loop.

Capability parity with vizier/_src/benchmarks/runners/benchmark_runner.py
(BenchmarkRunner :215, subroutines :75-213).
"""

from __future__ import annotations

import abc
import dataclasses
from typing import Optional, Sequence

from vizier_amd import pyvizier as vz
from vizier_amd._src.benchmarks.runners.benchmark_state import (
    BenchmarkState,
)


class BenchmarkSubroutine(abc.ABC):
  """One phase of a benchmark iteration."""

  @abc.abstractmethod
  def run(self, state: BenchmarkState) -> None:
    ...


@dataclasses.dataclass
class GenerateSuggestions(BenchmarkSubroutine):
  num_suggestions: int = 1

  def run(self, state: BenchmarkState) -> None:
    state.algorithm.suggest(self.num_suggestions)


@dataclasses.dataclass
class EvaluateActiveTrials(BenchmarkSubroutine):
  max_num_trials: Optional[int] = None

  def run(self, state: BenchmarkState) -> None:
    active = state.algorithm.supporter.GetTrials(
        status_matches=vz.TrialStatus.ACTIVE)
    if self.max_num_trials is not None:
      active = active[:self.max_num_trials]
    state.experimenter.evaluate(active)


@dataclasses.dataclass
class FillActiveTrials(BenchmarkSubroutine):
  """Tops up suggestions until `num_active_trials_limit` are ACTIVE
  (benchmark_runner.py:123)."""

  num_active_trials_limit: int = 1

  def run(self, state: BenchmarkState) -> None:
    active = state.algorithm.supporter.GetTrials(
        status_matches=vz.TrialStatus.ACTIVE)
    missing = self.num_active_trials_limit - len(active)
    if missing > 0:
      state.algorithm.suggest(missing)


@dataclasses.dataclass
class GenerateAndEvaluate(BenchmarkSubroutine):
  num_suggestions: int = 1

  def run(self, state: BenchmarkState) -> None:
    GenerateSuggestions(self.num_suggestions).run(state)
    EvaluateActiveTrials().run(state)


@dataclasses.dataclass
class BenchmarkRunner(BenchmarkSubroutine):
  """Repeats a list of subroutines `num_repeats` times."""

  benchmark_subroutines: Sequence[BenchmarkSubroutine]
  num_repeats: int = 1

  def run(self, state: BenchmarkState) -> None:
    for _ in range(self.num_repeats):
      for sub in self.benchmark_subroutines:
        sub.run(state)
