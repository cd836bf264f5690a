"""In-process server bundles.

Capability parity with vizier/_src/service/vizier_server.py:42,101
(DefaultVizierServer with in-process Pythia; DistributedPythiaVizierServer
with Pythia on a second port).
"""

from __future__ import annotations

import socket
from concurrent import futures
from typing import Optional

import grpc

from vizier_amd._src.service import constants, service_stubs, stubs_util
from vizier_amd._src.service.policy_factory import DefaultPolicyFactory
from vizier_amd._src.service.pythia_service import PythiaServicer
from vizier_amd._src.service.vizier_service import VizierServicer


def pick_unused_port() -> int:
  with socket.socket(socket.AF_INET, socket.SOCK_STREAM) as s:
    s.bind(('localhost', 0))
    return s.getsockname()[1]


class DefaultVizierServer:
  """Vizier gRPC server with an in-process Pythia servicer."""

  def __init__(self, host: str = 'localhost',
               database_url: Optional[str] = constants.SQL_MEMORY_URL,
               policy_factory=None, port: Optional[int] = None):
    self._host = host
    self._port = port or pick_unused_port()
    policy_factory = policy_factory or DefaultPolicyFactory()
    self._pythia_servicer = PythiaServicer(policy_factory=policy_factory)
    self._servicer = VizierServicer(
        database_url=database_url,
        default_pythia_service=self._pythia_servicer)
    self._pythia_servicer.connect_to_vizier(self._servicer)
    self._server = grpc.server(futures.ThreadPoolExecutor(max_workers=30))
    service_stubs.add_vizier_servicer_to_server(self._servicer, self._server)
    self._server.add_insecure_port(self.endpoint)
    self._server.start()

  @property
  def endpoint(self) -> str:
    return f'{self._host}:{self._port}'

  @property
  def servicer(self) -> VizierServicer:
    return self._servicer

  def stub(self) -> service_stubs.VizierServiceStub:
    return stubs_util.create_vizier_server_stub(self.endpoint)

  def stop(self, grace: Optional[float] = None):
    return self._server.stop(grace)

  def wait_for_termination(self, timeout: Optional[float] = None):
    return self._server.wait_for_termination(timeout)


class DistributedPythiaVizierServer(DefaultVizierServer):
  """Vizier server + a separate Pythia gRPC server wired over the wire."""

  def __init__(self, host: str = 'localhost',
               database_url: Optional[str] = constants.SQL_MEMORY_URL,
               policy_factory=None, port: Optional[int] = None,
               pythia_port: Optional[int] = None):
    self._host = host
    self._port = port or pick_unused_port()
    self._pythia_port = pythia_port or pick_unused_port()
    policy_factory = policy_factory or DefaultPolicyFactory()

    # Vizier server first (Pythia needs its endpoint for data access).
    self._servicer = VizierServicer(database_url=database_url,
                                    default_pythia_service=_LazyStub(self))
    self._server = grpc.server(futures.ThreadPoolExecutor(max_workers=30))
    service_stubs.add_vizier_servicer_to_server(self._servicer, self._server)
    self._server.add_insecure_port(self.endpoint)
    self._server.start()

    # Pythia server on its own port, talking back to Vizier over gRPC.
    vizier_stub = stubs_util.create_vizier_server_stub(self.endpoint)
    self._pythia_servicer = PythiaServicer(vizier_service=vizier_stub,
                                           policy_factory=policy_factory)
    # A single worker: Pythia computations are serialized like the reference.
    self._pythia_server = grpc.server(futures.ThreadPoolExecutor(
        max_workers=1))
    service_stubs.add_pythia_servicer_to_server(self._pythia_servicer,
                                                self._pythia_server)
    self._pythia_server.add_insecure_port(self.pythia_endpoint)
    self._pythia_server.start()

  @property
  def pythia_endpoint(self) -> str:
    return f'{self._host}:{self._pythia_port}'

  def stop(self, grace: Optional[float] = None):
    self._pythia_server.stop(grace)
    return self._server.stop(grace)


class _LazyStub:
  """Defers Pythia stub creation until the Pythia server is listening."""

  def __init__(self, server: 'DistributedPythiaVizierServer'):
    self._server = server

  def __getattr__(self, name):
    stub = stubs_util.create_pythia_server_stub(
        self._server.pythia_endpoint)
    return getattr(stub, name)
