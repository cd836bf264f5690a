"""Resource-name grammar for the service.

Wire/API parity with vizier/_src/service/resources.py:
  owners/{owner}
  owners/{owner}/studies/{study}
  owners/{owner}/studies/{study}/trials/{trial}
  owners/{owner}/operations/suggestion/{study}/{client}/{number}
  owners/{owner}/operations/earlystopping/{study}/{trial}
"""

from __future__ import annotations

import re
from typing import Union

_COMPONENT = r'[^\/]+'


def _validate(component: str, what: str) -> str:
  component = str(component)
  if not re.fullmatch(_COMPONENT, component):
    raise ValueError(f'Invalid {what}: {component!r}')
  return component


class OwnerResource:

  def __init__(self, owner_id: str):
    self._owner_id = _validate(owner_id, 'owner_id')

  @classmethod
  def from_name(cls, resource_name: str) -> 'OwnerResource':
    m = re.fullmatch(r'owners/(?P<owner>[^\/]+)', resource_name)
    if not m:
      raise ValueError(f'Incorrect resource name sent: {resource_name}')
    return cls(m.group('owner'))

  @property
  def owner_id(self) -> str:
    return self._owner_id

  @property
  def name(self) -> str:
    return f'owners/{self._owner_id}'


class StudyResource:

  def __init__(self, owner_id: str, study_id: str):
    self._owner_id = _validate(owner_id, 'owner_id')
    self._study_id = _validate(study_id, 'study_id')

  @classmethod
  def from_name(cls, resource_name: str) -> 'StudyResource':
    m = re.fullmatch(r'owners/(?P<owner>[^\/]+)/studies/(?P<study>[^\/]+)',
                     resource_name)
    if not m:
      raise ValueError(
          f'{resource_name!r} is not a valid name for a Study resource.')
    return cls(m.group('owner'), m.group('study'))

  @property
  def owner_id(self) -> str:
    return self._owner_id

  @property
  def study_id(self) -> str:
    return self._study_id

  @property
  def owner_resource(self) -> OwnerResource:
    return OwnerResource(self._owner_id)

  @property
  def name(self) -> str:
    return f'owners/{self._owner_id}/studies/{self._study_id}'

  def trial_resource(self, trial_id: Union[int, str]) -> 'TrialResource':
    return TrialResource(self._owner_id, self._study_id, int(trial_id))


class TrialResource:

  def __init__(self, owner_id: str, study_id: str, trial_id: int):
    self._owner_id = _validate(owner_id, 'owner_id')
    self._study_id = _validate(study_id, 'study_id')
    self._trial_id = int(trial_id)

  @classmethod
  def from_name(cls, resource_name: str) -> 'TrialResource':
    m = re.fullmatch(
        r'owners/(?P<owner>[^\/]+)/studies/(?P<study>[^\/]+)'
        r'/trials/(?P<trial>[^\/]+)', resource_name)
    if not m:
      raise ValueError(
          f'{resource_name!r} is not a valid name for a Trial resource.')
    return cls(m.group('owner'), m.group('study'), int(m.group('trial')))

  @property
  def owner_id(self) -> str:
    return self._owner_id

  @property
  def study_id(self) -> str:
    return self._study_id

  @property
  def trial_id(self) -> int:
    return self._trial_id

  @property
  def study_resource(self) -> StudyResource:
    return StudyResource(self._owner_id, self._study_id)

  @property
  def early_stopping_operation_resource(
      self) -> 'EarlyStoppingOperationResource':
    return EarlyStoppingOperationResource(self._owner_id, self._study_id,
                                          self._trial_id)

  @property
  def name(self) -> str:
    return (f'owners/{self._owner_id}/studies/{self._study_id}'
            f'/trials/{self._trial_id}')


class SuggestionOperationResource:

  def __init__(self, owner_id: str, study_id: str, client_id: str,
               operation_number: int):
    self._owner_id = _validate(owner_id, 'owner_id')
    self._study_id = _validate(study_id, 'study_id')
    self._client_id = _validate(client_id, 'client_id')
    self._operation_number = int(operation_number)

  @classmethod
  def from_name(cls, resource_name: str) -> 'SuggestionOperationResource':
    m = re.fullmatch(
        r'owners/(?P<owner>[^\/]+)/operations/suggestion/'
        r'(?P<study>[^\/]+)/(?P<client>[^\/]+)/(?P<number>[^\/]+)',
        resource_name)
    if not m:
      raise ValueError(f'Incorrect resource name sent: {resource_name}')
    return cls(m.group('owner'), m.group('study'), m.group('client'),
               int(m.group('number')))

  @property
  def owner_id(self) -> str:
    return self._owner_id

  @property
  def study_id(self) -> str:
    return self._study_id

  @property
  def client_id(self) -> str:
    return self._client_id

  @property
  def operation_number(self) -> int:
    return self._operation_number

  @property
  def operation_id(self) -> str:
    return f'suggestion/{self._study_id}/{self._client_id}/' \
        f'{self._operation_number}'

  @property
  def name(self) -> str:
    return (f'owners/{self._owner_id}/operations/suggestion/'
            f'{self._study_id}/{self._client_id}/{self._operation_number}')


class EarlyStoppingOperationResource:

  def __init__(self, owner_id: str, study_id: str, trial_id: int):
    self._owner_id = _validate(owner_id, 'owner_id')
    self._study_id = _validate(study_id, 'study_id')
    self._trial_id = int(trial_id)

  @classmethod
  def from_name(cls, resource_name: str) -> 'EarlyStoppingOperationResource':
    m = re.fullmatch(
        r'owners/(?P<owner>[^\/]+)/operations/earlystopping/'
        r'(?P<study>[^\/]+)/(?P<trial>[^\/]+)', resource_name)
    if not m:
      raise ValueError(f'Incorrect resource name sent: {resource_name}')
    return cls(m.group('owner'), m.group('study'), int(m.group('trial')))

  @property
  def owner_id(self) -> str:
    return self._owner_id

  @property
  def study_id(self) -> str:
    return self._study_id

  @property
  def trial_id(self) -> int:
    return self._trial_id

  @property
  def operation_id(self) -> str:
    return f'earlystopping/{self._study_id}/{self._trial_id}'

  @property
  def trial_resource(self) -> TrialResource:
    return TrialResource(self._owner_id, self._study_id, self._trial_id)

  @property
  def name(self) -> str:
    return (f'owners/{self._owner_id}/operations/earlystopping/'
            f'{self._study_id}/{self._trial_id}')
