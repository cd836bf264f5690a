"""Pythia servicer: runs policies on behalf of the Vizier service.

Capability parity with vizier/_src/service/pythia_service.py:36-125.
Works both in-process (called directly by VizierServicer) and as a gRPC
servicer via service_stubs.add_pythia_servicer_to_server.
"""

from __future__ import annotations

import logging
from typing import Optional

from google.protobuf import empty_pb2

from vizier_amd._src.service import grpc_util
from vizier_amd._src.service import pythia_converters
from vizier_amd._src.service.policy_factory import DefaultPolicyFactory
from vizier_amd._src.service.service_policy_supporter import (
    ServicePolicySupporter,
)
from vizier_amd._src.pythia.policy_factory import PolicyFactory

logger = logging.getLogger(__name__)


class PythiaServicer:
  """Receives Suggest/EarlyStop requests and runs the policy."""

  def __init__(self, vizier_service=None,
               policy_factory: Optional[PolicyFactory] = None):
    # Can be set after construction to break the circular dependency with
    # VizierServicer (which owns the default PythiaServicer).
    self._vizier_service = vizier_service
    self._policy_factory = policy_factory or DefaultPolicyFactory()

  def connect_to_vizier(self, vizier_service) -> None:
    if self._vizier_service is not None:
      raise ValueError('Vizier service was already set.')
    self._vizier_service = vizier_service

  def Suggest(self, request, context=None):
    try:
      py_request = pythia_converters.SuggestConverter.from_request_proto(
          request)
      supporter = ServicePolicySupporter(py_request.study_guid,
                                         self._vizier_service)
      policy = self._policy_factory(py_request.study_config.to_problem(),
                                    request.algorithm, supporter,
                                    py_request.study_guid)
      decision = policy.suggest(py_request)
      return pythia_converters.SuggestConverter.to_decision_proto(decision)
    except Exception as e:  # Surface as an RpcError to the caller.
      logger.exception('Pythia Suggest failed')
      if context is None:
        raise grpc_util.LocalRpcError(e) from e
      grpc_util.handle_exception(e, context)

  def EarlyStop(self, request, context=None):
    try:
      py_request = pythia_converters.EarlyStopConverter.from_request_proto(
          request)
      supporter = ServicePolicySupporter(py_request.study_guid,
                                         self._vizier_service)
      policy = self._policy_factory(py_request.study_config.to_problem(),
                                    request.algorithm, supporter,
                                    py_request.study_guid)
      decisions = policy.early_stop(py_request)
      return pythia_converters.EarlyStopConverter.to_decisions_proto(
          decisions)
    except Exception as e:
      logger.exception('Pythia EarlyStop failed')
      if context is None:
        raise grpc_util.LocalRpcError(e) from e
      grpc_util.handle_exception(e, context)

  def Ping(self, request, context=None):
    del request, context
    return empty_pb2.Empty()
