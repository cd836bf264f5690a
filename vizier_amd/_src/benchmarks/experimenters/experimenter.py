"""Experimenter ABC (parity with
vizier/_src/benchmarks/experimenters/experimenter.py:40)."""

from __future__ import annotations

import abc
from typing import Sequence

from vizier_amd import pyvizier as vz


class Experimenter(abc.ABC):
  """Evaluates suggested trials against a (synthetic) objective."""

  @abc.abstractmethod
  def evaluate(self, suggestions: Sequence[vz.Trial]) -> None:
    """Completes the given trials in place with measurements."""

  @abc.abstractmethod
  def problem_statement(self) -> vz.ProblemStatement:
    ...
