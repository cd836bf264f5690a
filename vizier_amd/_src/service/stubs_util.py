"""Channel/stub creation helpers.

Capability parity with vizier/_src/service/stubs_util.py:28-84 (insecure
channels with readiness wait and per-endpoint LRU-cached stubs).
"""

from __future__ import annotations

import functools

import grpc

from vizier_amd._src.service import service_stubs

_TIMEOUT_SECS = 10


def _create_channel(endpoint: str, timeout: float = _TIMEOUT_SECS
                    ) -> grpc.Channel:
  channel = grpc.insecure_channel(endpoint)
  grpc.channel_ready_future(channel).result(timeout=timeout)
  return channel


@functools.lru_cache(maxsize=None)
def create_vizier_server_stub(endpoint: str, timeout: float = _TIMEOUT_SECS
                              ) -> service_stubs.VizierServiceStub:
  return service_stubs.VizierServiceStub(_create_channel(endpoint, timeout))


@functools.lru_cache(maxsize=None)
def create_pythia_server_stub(endpoint: str, timeout: float = _TIMEOUT_SECS
                              ) -> service_stubs.PythiaServiceStub:
  return service_stubs.PythiaServiceStub(_create_channel(endpoint, timeout))
