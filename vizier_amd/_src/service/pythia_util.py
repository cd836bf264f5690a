"""Thread synchronization helper for cross-server Pythia calls.

Capability parity with vizier/_src/service/pythia_util.py:32 (ResponseWaiter).
"""

from __future__ import annotations

import threading
from typing import Generic, Optional, TypeVar

from vizier_amd._src.pythia import pythia_errors

_T = TypeVar('_T')


class ResponseWaiter(Generic[_T]):
  """A one-shot, thread-safe mailbox for an RPC response."""

  def __init__(self):
    self._lock = threading.Lock()
    self._event = threading.Event()
    self._response: Optional[_T] = None
    self._error: Optional[Exception] = None

  def Report(self, response: _T) -> None:
    with self._lock:
      if self._event.is_set():
        raise pythia_errors.PythiaProtocolError(
            'Response was already reported.')
      self._response = response
      self._event.set()

  def ReportError(self, error: Exception) -> None:
    with self._lock:
      self._error = error
      self._event.set()

  def WaitForResponse(self, timeout: Optional[float] = None) -> _T:
    if not self._event.wait(timeout):
      raise pythia_errors.PythiaProtocolError('Timed out waiting for '
                                              'response.')
    with self._lock:
      if self._error is not None:
        raise self._error
      return self._response
