"""gRPC helpers (parity with vizier/_src/service/grpc_util.py)."""

from __future__ import annotations

from typing import Optional

import grpc


class LocalRpcError(grpc.RpcError):
  """RpcError raised by in-process servicers (no socket involved)."""

  def __init__(self, exception: Exception,
               code: grpc.StatusCode = grpc.StatusCode.INTERNAL):
    super().__init__(str(exception))
    self._code = code
    self._details = f'{type(exception).__name__}: {exception}'

  def code(self) -> grpc.StatusCode:
    return self._code

  def details(self) -> str:
    return self._details


def _status_code(e: Exception) -> grpc.StatusCode:
  # Local import to avoid a cycle at module load.
  from vizier_amd._src.service import custom_errors
  if isinstance(e, (custom_errors.ImmutableStudyError,
                    custom_errors.ImmutableTrialError)):
    return grpc.StatusCode.FAILED_PRECONDITION
  if isinstance(e, custom_errors.NotFoundError):
    return grpc.StatusCode.NOT_FOUND
  return grpc.StatusCode.INTERNAL


def handle_exception(e: Exception,
                     context: Optional[grpc.ServicerContext] = None) -> None:
  """Aborts the RPC (gRPC path) or raises (in-process path)."""
  if context is None:
    raise e
  context.abort(_status_code(e), f'{type(e).__name__}: {e}')
