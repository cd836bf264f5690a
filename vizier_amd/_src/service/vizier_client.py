"""Low-level client with suggestion-operation polling.

Capability parity with vizier/_src/service/vizier_client.py (VizierClient
:94, create_or_load_study :417, PollingDelay :468, global
environment_variables :47-91 with NO_ENDPOINT -> in-process servicer).
"""

from __future__ import annotations

import datetime
import functools
import logging
import time
from typing import Any, Dict, List, Mapping, Optional, Union

import grpc

from vizier_amd._src.pyvizier import proto_converters as pc
from vizier_amd._src.pyvizier.study_config import StudyConfig
from vizier_amd._src.pyvizier.trial import Measurement, Trial
from vizier_amd._src.service import constants, custom_errors, resources
from vizier_amd._src.service import stubs_util
from vizier_amd._src.service.proto import (
    operations_pb2,
    study_pb2,
    vizier_service_pb2,
)

logger = logging.getLogger(__name__)


class _EnvironmentVariables:
  """Global client configuration (mirrors vizier_client.py:47)."""

  def __init__(self):
    self.server_endpoint: str = constants.NO_ENDPOINT
    self.servicer_kwargs: Dict[str, Any] = {}
    self.new_suggestion_polling_secs: float = 1.0

  def servicer_use_sql_ram(self) -> None:
    self.servicer_kwargs['database_url'] = constants.SQL_MEMORY_URL


environment_variables = _EnvironmentVariables()


@functools.lru_cache(maxsize=None)
def _create_local_vizier_servicer():
  from vizier_amd._src.service import vizier_service
  return vizier_service.VizierServicer(
      **environment_variables.servicer_kwargs)


def create_vizier_servicer_or_stub():
  endpoint = environment_variables.server_endpoint
  if endpoint == constants.NO_ENDPOINT:
    return _create_local_vizier_servicer()
  return stubs_util.create_vizier_server_stub(endpoint)


def PollingDelay(num_attempts: int, time_scale: float) -> datetime.timedelta:
  """Bounded exponential backoff starting at `time_scale` seconds."""
  small_interval = 0.3
  interval = max(time_scale, small_interval) * 1.41 ** min(num_attempts, 9)
  return datetime.timedelta(seconds=interval)


class VizierClient:
  """Talks to a VizierService (in-process servicer or gRPC stub)."""

  def __init__(self, study_resource_name: str, client_id: str,
               service=None):
    if not client_id:
      raise ValueError('client_id must be non-empty.')
    self._study_resource_name = study_resource_name
    self._client_id = client_id
    self._service = service if service is not None \
        else create_vizier_servicer_or_stub()

  @property
  def _study_resource(self) -> resources.StudyResource:
    return resources.StudyResource.from_name(self._study_resource_name)

  @property
  def _owner_id(self) -> str:
    return self._study_resource.owner_id

  @property
  def _study_id(self) -> str:
    return self._study_resource.study_id

  @property
  def study_resource_name(self) -> str:
    return self._study_resource_name

  def _trial_name(self, trial_id: int) -> str:
    return resources.TrialResource(self._owner_id, self._study_id,
                                   trial_id).name

  # -- suggestions ----------------------------------------------------------

  def get_suggestions(self, suggestion_count: int, *,
                      client_id_override: Optional[str] = None
                      ) -> List[Trial]:
    client_id = client_id_override or self._client_id
    request = vizier_service_pb2.SuggestTrialsRequest(
        parent=self._study_resource.name,
        suggestion_count=suggestion_count, client_id=client_id)
    try:
      operation = self._service.SuggestTrials(request)
    except grpc.RpcError as e:
      if e.code() == grpc.StatusCode.FAILED_PRECONDITION:
        return []
      raise
    except custom_errors.ImmutableStudyError:
      return []

    num_attempts = 0
    while not operation.done:
      delay = PollingDelay(num_attempts,
                           environment_variables.new_suggestion_polling_secs)
      num_attempts += 1
      time.sleep(delay.total_seconds())
      operation = self._service.GetOperation(
          operations_pb2.GetOperationRequest(name=operation.name))

    if operation.HasField('error'):
      raise RuntimeError(f'SuggestOperation {operation.name} failed: '
                         f'{operation.error}')
    trials = vizier_service_pb2.SuggestTrialsResponse.FromString(
        operation.response.value).trials
    return pc.TrialConverter.from_protos(trials)

  # -- measurements & completion -------------------------------------------

  def report_intermediate_objective_value(
      self, step: int, elapsed_secs: float,
      metric_list: List[Mapping[str, Union[int, float]]],
      trial_id: int) -> Trial:
    metrics = {}
    for entry in metric_list:
      for name, value in entry.items():
        metrics[name] = float(value)
    measurement = Measurement(metrics=metrics, elapsed_secs=elapsed_secs,
                              steps=step)
    request = vizier_service_pb2.AddTrialMeasurementRequest(
        trial_name=self._trial_name(trial_id))
    request.measurement.CopyFrom(pc.MeasurementConverter.to_proto(
        measurement))
    trial = self._service.AddTrialMeasurement(request)
    return pc.TrialConverter.from_proto(trial)

  def should_trial_stop(self, trial_id: int) -> bool:
    request = vizier_service_pb2.CheckTrialEarlyStoppingStateRequest(
        trial_name=self._trial_name(trial_id))
    resp = self._service.CheckTrialEarlyStoppingState(request)
    return resp.should_stop

  def stop_trial(self, trial_id: int) -> None:
    request = vizier_service_pb2.StopTrialRequest(
        name=self._trial_name(trial_id))
    self._service.StopTrial(request)

  def complete_trial(self, trial_id: int,
                     final_measurement: Optional[Measurement] = None,
                     infeasibility_reason: Optional[str] = None) -> Trial:
    request = vizier_service_pb2.CompleteTrialRequest(
        name=self._trial_name(trial_id),
        trial_infeasible=infeasibility_reason is not None,
        infeasible_reason=infeasibility_reason or '')
    if final_measurement is not None:
      request.final_measurement.CopyFrom(
          pc.MeasurementConverter.to_proto(final_measurement))
    trial = self._service.CompleteTrial(request)
    return pc.TrialConverter.from_proto(trial)

  # -- reads ----------------------------------------------------------------

  def get_trial(self, trial_id: int) -> Trial:
    trial = self._service.GetTrial(vizier_service_pb2.GetTrialRequest(
        name=self._trial_name(trial_id)))
    return pc.TrialConverter.from_proto(trial)

  def list_trials(self) -> List[Trial]:
    resp = self._service.ListTrials(vizier_service_pb2.ListTrialsRequest(
        parent=self._study_resource.name))
    return pc.TrialConverter.from_protos(resp.trials)

  def list_optimal_trials(self) -> List[Trial]:
    resp = self._service.ListOptimalTrials(
        vizier_service_pb2.ListOptimalTrialsRequest(
            parent=self._study_resource.name))
    return pc.TrialConverter.from_protos(resp.optimal_trials)

  def list_studies(self) -> List[Dict[str, Any]]:
    resp = self._service.ListStudies(vizier_service_pb2.ListStudiesRequest(
        parent=resources.OwnerResource(self._owner_id).name))
    return [{'name': s.name, 'display_name': s.display_name}
            for s in resp.studies]

  # -- trial/study mutations ------------------------------------------------

  def add_trial(self, trial: Trial) -> Trial:
    request = vizier_service_pb2.CreateTrialRequest(
        parent=self._study_resource.name)
    request.trial.CopyFrom(pc.TrialConverter.to_proto(trial))
    out = self._service.CreateTrial(request)
    return pc.TrialConverter.from_proto(out)

  def delete_trial(self, trial_id: int) -> None:
    self._service.DeleteTrial(vizier_service_pb2.DeleteTrialRequest(
        name=self._trial_name(trial_id)))

  def delete_study(self, study_resource_name: Optional[str] = None) -> None:
    name = study_resource_name or self._study_resource.name
    self._service.DeleteStudy(vizier_service_pb2.DeleteStudyRequest(
        name=name))

  def get_study_config(self, study_name: Optional[str] = None) -> StudyConfig:
    name = study_name or self._study_resource.name
    study = self._service.GetStudy(vizier_service_pb2.GetStudyRequest(
        name=name))
    return StudyConfig.from_proto(study.study_spec)

  def get_study_state(self, study_name: Optional[str] = None):
    name = study_name or self._study_resource.name
    study = self._service.GetStudy(vizier_service_pb2.GetStudyRequest(
        name=name))
    return study.state

  def set_study_state(self, state, study_name: Optional[str] = None):
    name = study_name or self._study_resource.name
    study = self._service.SetStudyState(
        vizier_service_pb2.SetStudyStateRequest(parent=name, state=state))
    return study.state

  def update_metadata(self, delta) -> None:
    """Applies a MetadataDelta to the study (bulk atomic update)."""
    from vizier_amd._src.pyvizier import metadata_util
    request = vizier_service_pb2.UpdateMetadataRequest(
        name=self._study_resource.name)
    for u in metadata_util.study_metadata_to_update_list(delta.on_study):
      request.delta.add().CopyFrom(u)
    for u in metadata_util.trial_metadata_to_update_list(delta.on_trials):
      request.delta.add().CopyFrom(u)
    resp = self._service.UpdateMetadata(request)
    if resp.error_details:
      raise KeyError(resp.error_details)


def create_or_load_study(owner_id: str, client_id: str, study_id: str,
                         study_config: StudyConfig) -> VizierClient:
  """Creates the study (idempotently) and returns a client bound to it."""
  service = create_vizier_servicer_or_stub()
  study = study_pb2.Study(display_name=study_id)
  study.study_spec.CopyFrom(study_config.to_proto())
  request = vizier_service_pb2.CreateStudyRequest(
      parent=resources.OwnerResource(owner_id).name, study=study)
  study = service.CreateStudy(request)
  return VizierClient(study.name, client_id, service)
