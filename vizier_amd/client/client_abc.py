"""Abstract cross-platform client interfaces.

Capability parity with vizier/client/client_abc.py (TrialInterface :47,
StudyInterface :191).
"""

from __future__ import annotations

import abc
from typing import Any, Callable, Iterator, Mapping, Optional, Type, TypeVar

from vizier_amd import pyvizier as vz

_T = TypeVar('_T')


class ResourceNotFoundError(LookupError):
  """Raised when a study/trial resource does not exist."""


class TrialInterface(abc.ABC):
  """A trial in a study, with mutation/read methods."""

  @property
  @abc.abstractmethod
  def id(self) -> int:
    ...

  @property
  @abc.abstractmethod
  def parameters(self) -> Mapping[str, Any]:
    ...

  @abc.abstractmethod
  def delete(self) -> None:
    ...

  @abc.abstractmethod
  def update_metadata(self, delta: vz.Metadata) -> None:
    ...

  @abc.abstractmethod
  def complete(self, measurement: Optional[vz.Measurement] = None, *,
               infeasible_reason: Optional[str] = None
               ) -> Optional[vz.Measurement]:
    ...

  @abc.abstractmethod
  def check_early_stopping(self) -> bool:
    ...

  @abc.abstractmethod
  def stop(self) -> None:
    ...

  @abc.abstractmethod
  def add_measurement(self, measurement: vz.Measurement) -> None:
    ...

  @abc.abstractmethod
  def materialize(self, *, include_all_measurements: bool = True) -> vz.Trial:
    ...

  @property
  @abc.abstractmethod
  def study(self) -> 'StudyInterface':
    ...


class TrialIterable(abc.ABC):
  """Iterates TrialInterfaces with a shortcut for materialized trials."""

  @abc.abstractmethod
  def __iter__(self) -> Iterator[TrialInterface]:
    ...

  @abc.abstractmethod
  def get(self) -> Iterator[vz.Trial]:
    ...


class StudyInterface(abc.ABC):
  """A study: suggestion source and trial container."""

  @property
  @abc.abstractmethod
  def resource_name(self) -> str:
    ...

  @abc.abstractmethod
  def suggest(self, *, count: Optional[int] = None,
              client_id: str = 'default_client_id'):
    ...

  @abc.abstractmethod
  def request(self, suggestion: vz.TrialSuggestion) -> TrialInterface:
    ...

  @abc.abstractmethod
  def delete(self) -> None:
    ...

  @abc.abstractmethod
  def update_metadata(self, delta: vz.Metadata) -> None:
    ...

  @abc.abstractmethod
  def add_trial(self, trial: vz.Trial) -> TrialInterface:
    ...

  @abc.abstractmethod
  def trials(self, trial_filter: Optional[vz.TrialFilter] = None
             ) -> TrialIterable:
    ...

  @abc.abstractmethod
  def get_trial(self, uid: int) -> TrialInterface:
    ...

  @abc.abstractmethod
  def optimal_trials(self, *, count: Optional[int] = None) -> TrialIterable:
    ...

  @abc.abstractmethod
  def materialize_problem_statement(self) -> vz.ProblemStatement:
    ...

  @abc.abstractmethod
  def set_state(self, state: vz.StudyState) -> None:
    ...

  @abc.abstractmethod
  def materialize_state(self) -> vz.StudyState:
    ...

  @classmethod
  @abc.abstractmethod
  def from_resource_name(cls: Type[_T], name: str) -> _T:
    ...
