"""Datastore contract for studies, trials and operations.

Capability parity with vizier/_src/service/datastore.py:34-244. All
implementations are pass-by-value: protos returned are copies, and stored
protos are copies of the arguments.
"""

from __future__ import annotations

import abc
from typing import Callable, Iterable, List, Optional


class DataStore(abc.ABC):
  """Abstract storage for studies, trials, and long-running operations."""

  # -- studies --------------------------------------------------------------

  @abc.abstractmethod
  def create_study(self, study) -> str:
    """Stores a Study proto; returns its resource name. Raises
    AlreadyExistsError on duplicates."""

  @abc.abstractmethod
  def load_study(self, study_name: str):
    """Returns the Study proto. Raises NotFoundError."""

  @abc.abstractmethod
  def update_study(self, study) -> str:
    """Replaces an existing study. Raises NotFoundError."""

  @abc.abstractmethod
  def delete_study(self, study_name: str) -> None:
    """Deletes the study and all of its trials/operations."""

  @abc.abstractmethod
  def list_studies(self, owner_name: str) -> List:
    """All studies of an owner. Raises NotFoundError for unknown owners."""

  # -- trials ---------------------------------------------------------------

  @abc.abstractmethod
  def create_trial(self, trial) -> str:
    """Stores a Trial proto (trial.name determines the study)."""

  @abc.abstractmethod
  def get_trial(self, trial_name: str):
    ...

  @abc.abstractmethod
  def update_trial(self, trial) -> str:
    ...

  @abc.abstractmethod
  def list_trials(self, study_name: str) -> List:
    ...

  @abc.abstractmethod
  def delete_trial(self, trial_name: str) -> None:
    ...

  @abc.abstractmethod
  def max_trial_id(self, study_name: str) -> int:
    """Largest trial id in the study (0 if none)."""

  # -- suggestion operations ------------------------------------------------

  @abc.abstractmethod
  def create_suggestion_operation(self, operation) -> str:
    ...

  @abc.abstractmethod
  def get_suggestion_operation(self, operation_name: str):
    ...

  @abc.abstractmethod
  def update_suggestion_operation(self, operation) -> str:
    ...

  @abc.abstractmethod
  def list_suggestion_operations(
      self, study_name: str, client_id: str,
      filter_fn: Optional[Callable[[object], bool]] = None) -> List:
    ...

  @abc.abstractmethod
  def max_suggestion_operation_number(self, study_name: str,
                                      client_id: str) -> int:
    ...

  # -- early-stopping operations -------------------------------------------

  @abc.abstractmethod
  def create_early_stopping_operation(self, operation) -> str:
    ...

  @abc.abstractmethod
  def get_early_stopping_operation(self, operation_name: str):
    ...

  @abc.abstractmethod
  def update_early_stopping_operation(self, operation) -> str:
    ...

  # -- metadata -------------------------------------------------------------

  @abc.abstractmethod
  def update_metadata(self, study_name: str, study_metadata: Iterable,
                      trial_metadata: Iterable) -> None:
    """Writes study metadata (KeyValue protos) and trial metadata
    (UnitMetadataUpdate protos). Raises KeyError on unknown study/trial."""
