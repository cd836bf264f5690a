"""ProblemAndTrials bundle (parity with
vizier/_src/pyvizier/shared/study.py:26)."""

from __future__ import annotations

from typing import Iterable, List

from vizier_amd._src.pyvizier.base_study_config import ProblemStatement
from vizier_amd._src.pyvizier.trial import Trial


class ProblemAndTrials:
  """A problem statement together with a list of trials."""

  def __init__(self, problem: ProblemStatement,
               trials: Iterable[Trial] = ()):
    self.problem = problem
    self.trials: List[Trial] = list(trials)

  def __eq__(self, other) -> bool:
    if not isinstance(other, ProblemAndTrials):
      return NotImplemented
    return self.problem == other.problem and self.trials == other.trials

  def __repr__(self) -> str:
    return (f'ProblemAndTrials(problem={self.problem!r}, '
            f'n_trials={len(self.trials)})')
