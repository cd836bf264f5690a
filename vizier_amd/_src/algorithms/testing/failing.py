"""Fault-injection designers for testing.

Capability parity with vizier/_src/algorithms/testing/failing.py
(FailedSuggestError :25, FailingDesigner :29,
AlternateFailingDesigner :46): used to exercise the service's
Pythia-error capture path (errors land in the suggest operation's
status and surface to polling clients) and ensemble robustness.
"""

from __future__ import annotations

from typing import Optional, Sequence

from vizier_amd import pyvizier as vz
from vizier_amd._src.algorithms.core.abstractions import (
    ActiveTrials,
    CompletedTrials,
    Designer,
)
from vizier_amd._src.algorithms.designers.random import RandomDesigner


class FailedSuggestError(Exception):
  """Raised by the failing designers' suggest calls."""


class FailingDesigner(Designer):
  """Raises at every suggest call."""

  def update(self, completed: CompletedTrials,
             all_active: ActiveTrials) -> None:
    pass

  def suggest(self, count: Optional[int] = None
              ) -> Sequence[vz.TrialSuggestion]:
    raise FailedSuggestError()


class AlternateFailingDesigner(Designer):
  """Raises at every second suggest call (in-memory runs only)."""

  def __init__(self, search_space: vz.SearchSpace):
    self._suggest_count = 0
    self._random_designer = RandomDesigner(search_space)

  def update(self, completed: CompletedTrials,
             all_active: ActiveTrials) -> None:
    pass

  def suggest(self, count: Optional[int] = None
              ) -> Sequence[vz.TrialSuggestion]:
    count = count or 1
    self._suggest_count += count
    if self._suggest_count % 2 == 0 or count > 1:
      raise FailedSuggestError()
    return self._random_designer.suggest(1)
