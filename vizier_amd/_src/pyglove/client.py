"""VizierTuner abstraction for the pyglove backend.

Parity with vizier/_src/pyglove/client.py: the per-platform tuner
(OSS / Vertex / internal) behind VizierBackend — study CRUD, Pythia
hosting, chief liveness pings.
"""

from __future__ import annotations

import abc
import threading
from typing import Dict, NewType, Optional, Union

from vizier_amd import pyvizier as vz

ExpandedStudyName = NewType('ExpandedStudyName', str)


class StudyKey:
  """Immutable (owner, name) key for the policy cache."""

  def __init__(self, owner: str, name: str):
    self._key = (owner, str(name))

  def __hash__(self):
    return hash(self._key)

  def __eq__(self, other):
    return isinstance(other, StudyKey) and other._key == self._key

  def __repr__(self):
    return f'StudyKey{self._key}'


PolicyCache = Dict[StudyKey, object]


class VizierTuner(abc.ABC):
  """Platform-specific tuner operations (client.py:49)."""

  def __init__(self):
    self._pythia_lock = threading.Lock()

  @abc.abstractmethod
  def get_tuner_id(self, algorithm) -> str:
    ...

  def start_pythia_service(self, policy_cache: PolicyCache) -> None:
    with self._pythia_lock:
      self._start_pythia_service(policy_cache)

  @abc.abstractmethod
  def _start_pythia_service(self, policy_cache: PolicyCache) -> None:
    ...

  @abc.abstractmethod
  def load_prior_study(self, resource_name: str):
    ...

  @abc.abstractmethod
  def create_study(self, problem: vz.ProblemStatement, converter,
                   owner: str, name: str, algorithm,
                   stopping_policy=None):
    ...

  @abc.abstractmethod
  def get_group_id(self, group_id: Union[None, int, str] = None) -> str:
    ...

  @abc.abstractmethod
  def ping_tuner(self, tuner_id: str) -> bool:
    ...

  @abc.abstractmethod
  def pythia_supporter(self, study):
    ...

  @abc.abstractmethod
  def use_pythia_for_study(self, study) -> None:
    ...

  @classmethod
  @abc.abstractmethod
  def load_study(cls, owner: str, name: ExpandedStudyName):
    ...
