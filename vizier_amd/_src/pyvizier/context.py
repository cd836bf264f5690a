"""Context object (capability parity with
vizier/_src/pyvizier/shared/context.py)."""

from __future__ import annotations

from typing import Any, Mapping, Optional

from vizier_amd._src.pyvizier.common import Metadata
from vizier_amd._src.pyvizier.trial import ParameterDict


class Context:
  """Extra information attached to a study evaluation context."""

  def __init__(self, parameters: Optional[Mapping[str, Any]] = None, *,
               metadata: Optional[Metadata] = None,
               description: Optional[str] = None):
    self.parameters = (parameters if isinstance(parameters, ParameterDict)
                       else ParameterDict(parameters or {}))
    self.metadata = metadata if metadata is not None else Metadata()
    self.description = description

  def __eq__(self, other) -> bool:
    if not isinstance(other, Context):
      return NotImplemented
    return (self.parameters == other.parameters and
            self.metadata == other.metadata and
            self.description == other.description)

  def __repr__(self) -> str:
    return f'Context(parameters={self.parameters!r})'
