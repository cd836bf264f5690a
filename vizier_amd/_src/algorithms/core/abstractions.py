"""Designer abstractions.

Capability parity with vizier/_src/algorithms/core/abstractions.py
(Designer :92-150, Predictor :152-199, (Partially)SerializableDesigner
:209-215).
"""

from __future__ import annotations

import abc
import dataclasses
from typing import Callable, List, Optional, Sequence, TypeVar

import numpy as np

from vizier_amd._src.pyvizier.common import Metadata
from vizier_amd._src.pyvizier.trial import Trial, TrialSuggestion


@dataclasses.dataclass(frozen=True)
class CompletedTrials:
  """Completed trials delivered to Designer.update exactly once."""

  trials: Sequence[Trial] = ()

  def __post_init__(self):
    object.__setattr__(self, 'trials', tuple(self.trials))

  @property
  def completed(self) -> Sequence[Trial]:
    return self.trials


@dataclasses.dataclass(frozen=True)
class ActiveTrials:
  """All currently-ACTIVE trials, delivered on every update."""

  trials: Sequence[Trial] = ()

  def __post_init__(self):
    object.__setattr__(self, 'trials', tuple(self.trials))


@dataclasses.dataclass
class Prediction:
  """Mean/stddev predictions with optional metadata."""

  mean: np.ndarray
  stddev: np.ndarray
  metadata: Optional[Metadata] = None


_T = TypeVar('_T')
# Factory protocol: problem statement (+kwargs) -> designer instance.
DesignerFactory = Callable[..., _T]


class Designer(abc.ABC):
  """A suggestion algorithm with incremental state updates."""

  @abc.abstractmethod
  def update(self, completed: CompletedTrials, all_active: ActiveTrials
             ) -> None:
    """Incorporates newly completed trials and the current active set."""

  @abc.abstractmethod
  def suggest(self, count: Optional[int] = None) -> Sequence[TrialSuggestion]:
    """Returns up to `count` suggestions (may under- or over-deliver)."""


class Predictor(abc.ABC):
  """Mixin for designers that can predict on unseen trials."""

  @abc.abstractmethod
  def predict(self, trials: Sequence[TrialSuggestion],
              rng: Optional[np.random.Generator] = None,
              num_samples: Optional[int] = None) -> Prediction:
    ...


class SerializableDesigner(Designer):
  """Designer whose full state round-trips through Metadata."""

  @abc.abstractmethod
  def dump(self) -> Metadata:
    ...

  @classmethod
  @abc.abstractmethod
  def recover(cls, metadata: Metadata) -> 'SerializableDesigner':
    ...


class PartiallySerializableDesigner(Designer):
  """Designer that can save/restore state given its constructor args."""

  @abc.abstractmethod
  def dump(self) -> Metadata:
    ...

  @abc.abstractmethod
  def load(self, metadata: Metadata) -> None:
    ...
