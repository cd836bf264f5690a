"""gRPC service registration + client stubs, built without grpc_tools.

Provides the same method paths as generated `*_pb2_grpc` code
(/vizier.VizierService/<Method>, /vizier.PythiaService/<Method>) so the
wire protocol matches the reference service.
"""

from __future__ import annotations

import grpc
from google.protobuf import empty_pb2

from vizier_amd._src.service.proto import (
    operations_pb2,
    pythia_service_pb2,
    study_pb2,
    vizier_service_pb2,
)

VIZIER_SERVICE_NAME = 'vizier.VizierService'
PYTHIA_SERVICE_NAME = 'vizier.PythiaService'

# method -> (request class, response class)
VIZIER_METHODS = {
    'CreateStudy': (vizier_service_pb2.CreateStudyRequest, study_pb2.Study),
    'GetStudy': (vizier_service_pb2.GetStudyRequest, study_pb2.Study),
    'ListStudies': (vizier_service_pb2.ListStudiesRequest,
                    vizier_service_pb2.ListStudiesResponse),
    'DeleteStudy': (vizier_service_pb2.DeleteStudyRequest, empty_pb2.Empty),
    'SetStudyState': (vizier_service_pb2.SetStudyStateRequest,
                      study_pb2.Study),
    'SuggestTrials': (vizier_service_pb2.SuggestTrialsRequest,
                      operations_pb2.Operation),
    'GetOperation': (operations_pb2.GetOperationRequest,
                     operations_pb2.Operation),
    'CreateTrial': (vizier_service_pb2.CreateTrialRequest, study_pb2.Trial),
    'GetTrial': (vizier_service_pb2.GetTrialRequest, study_pb2.Trial),
    'ListTrials': (vizier_service_pb2.ListTrialsRequest,
                   vizier_service_pb2.ListTrialsResponse),
    'AddTrialMeasurement': (vizier_service_pb2.AddTrialMeasurementRequest,
                            study_pb2.Trial),
    'CompleteTrial': (vizier_service_pb2.CompleteTrialRequest,
                      study_pb2.Trial),
    'DeleteTrial': (vizier_service_pb2.DeleteTrialRequest, empty_pb2.Empty),
    'CheckTrialEarlyStoppingState': (
        vizier_service_pb2.CheckTrialEarlyStoppingStateRequest,
        vizier_service_pb2.CheckTrialEarlyStoppingStateResponse),
    'StopTrial': (vizier_service_pb2.StopTrialRequest, study_pb2.Trial),
    'ListOptimalTrials': (vizier_service_pb2.ListOptimalTrialsRequest,
                          vizier_service_pb2.ListOptimalTrialsResponse),
    'UpdateMetadata': (vizier_service_pb2.UpdateMetadataRequest,
                       vizier_service_pb2.UpdateMetadataResponse),
}

PYTHIA_METHODS = {
    'Suggest': (pythia_service_pb2.SuggestRequest,
                pythia_service_pb2.SuggestDecision),
    'EarlyStop': (pythia_service_pb2.EarlyStopRequest,
                  pythia_service_pb2.EarlyStopDecisions),
    'Ping': (empty_pb2.Empty, empty_pb2.Empty),
}


def _add_servicer(servicer, server, service_name, methods) -> None:
  handlers = {}
  for name, (req_cls, _resp_cls) in methods.items():
    handlers[name] = grpc.unary_unary_rpc_method_handler(
        getattr(servicer, name),
        request_deserializer=req_cls.FromString,
        response_serializer=lambda msg: msg.SerializeToString())
  server.add_generic_rpc_handlers(
      (grpc.method_handlers_generic_handler(service_name, handlers),))


def add_vizier_servicer_to_server(servicer, server) -> None:
  _add_servicer(servicer, server, VIZIER_SERVICE_NAME, VIZIER_METHODS)


def add_pythia_servicer_to_server(servicer, server) -> None:
  _add_servicer(servicer, server, PYTHIA_SERVICE_NAME, PYTHIA_METHODS)


class _Stub:

  def __init__(self, channel: grpc.Channel, service_name: str, methods):
    for name, (req_cls, resp_cls) in methods.items():
      setattr(self, name, channel.unary_unary(
          f'/{service_name}/{name}',
          request_serializer=lambda msg: msg.SerializeToString(),
          response_deserializer=resp_cls.FromString))


class VizierServiceStub(_Stub):

  def __init__(self, channel: grpc.Channel):
    super().__init__(channel, VIZIER_SERVICE_NAME, VIZIER_METHODS)


class PythiaServiceStub(_Stub):

  def __init__(self, channel: grpc.Channel):
    super().__init__(channel, PYTHIA_SERVICE_NAME, PYTHIA_METHODS)
