"""(Partially)Serializable interfaces.

Capability parity with vizier/interfaces/serializable.py:27-108.
"""

from __future__ import annotations

import abc

from vizier_amd._src.pyvizier.common import Metadata


class DecodeError(Exception):
  """Base error for load/recover failures."""


class HarmlessDecodeError(DecodeError):
  """State could not be read, but a fresh object is safe to use."""


class FatalDecodeError(DecodeError):
  """State is corrupt in a way that must not be silently ignored."""


class Serializable(abc.ABC):
  """Objects whose full state round-trips through Metadata."""

  @classmethod
  @abc.abstractmethod
  def recover(cls, metadata: Metadata) -> 'Serializable':
    """Raises HarmlessDecodeError/FatalDecodeError on bad state."""

  @abc.abstractmethod
  def dump(self) -> Metadata:
    ...


class PartiallySerializable(abc.ABC):
  """Objects that restore state given their constructor arguments."""

  @abc.abstractmethod
  def load(self, metadata: Metadata) -> None:
    """Raises HarmlessDecodeError/FatalDecodeError on bad state."""

  @abc.abstractmethod
  def dump(self) -> Metadata:
    ...
