"""Built-in algorithm markers for the pyglove backend.

Parity with vizier/_src/pyglove/algorithms.py: a `BuiltinAlgorithm` is
a pg.DNAGenerator-shaped marker telling the backend to run one of the
service's own algorithm strings instead of hosting a pyglove algorithm
behind Pythia.
"""

from __future__ import annotations

import pyglove as pg


class PseudoAlgorithm(pg.DNAGenerator):
  """Algorithms that the Vizier service runs natively (no Pythia host)."""


class BuiltinAlgorithm(PseudoAlgorithm):
  """Named Vizier-builtin algorithm, e.g. 'GAUSSIAN_PROCESS_BANDIT'."""

  def __init__(self, name: str):
    super().__init__()
    self._name = name

  @property
  def name(self) -> str:
    return self._name

  def _propose(self):
    raise NotImplementedError(
        f'{self._name} runs inside the Vizier service, not in-process.')

  def __eq__(self, other) -> bool:
    return (isinstance(other, BuiltinAlgorithm) and
            other.name == self._name)

  def __repr__(self) -> str:
    return f'BuiltinAlgorithm({self._name!r})'
