"""Automated (early) stopping configuration.

Capability parity with vizier/_src/pyvizier/oss/automated_stopping.py:29.
"""

from __future__ import annotations

from vizier_amd._src.service.proto import study_pb2


class AutomatedStoppingConfig:
  """Wrapper around StudySpec.DefaultEarlyStoppingSpec."""

  def __init__(self, proto):
    self._proto = proto

  @classmethod
  def default_stopping_spec(cls) -> 'AutomatedStoppingConfig':
    return cls(study_pb2.StudySpec.DefaultEarlyStoppingSpec())

  @classmethod
  def from_proto(cls, proto) -> 'AutomatedStoppingConfig':
    return cls(proto)

  def to_proto(self):
    return self._proto

  def __eq__(self, other) -> bool:
    if not isinstance(other, AutomatedStoppingConfig):
      return NotImplemented
    return self._proto == other._proto


from vizier_amd._src.service.proto import study_pb2 as _study_pb2

AutomatedStoppingConfigProto = _study_pb2.StudySpec.DefaultEarlyStoppingSpec
