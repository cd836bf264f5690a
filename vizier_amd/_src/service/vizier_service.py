"""The Vizier service brain: study/trial CRUD + suggestion state machine.

Capability parity with vizier/_src/service/vizier_service.py:64
(VizierServicer): per-study operation locks, SuggestTrials source order
(active-op dedup -> client's ACTIVE trials -> REQUESTED pool -> Pythia,
with over-delivery banked as REQUESTED), early-stopping operation
recycling, CompleteTrial final-measurement auto-selection, and
ListOptimalTrials Pareto filtering.
"""

from __future__ import annotations

import collections
import datetime
import logging
import threading
from typing import Optional

import grpc
import numpy as np
from google.protobuf import empty_pb2, timestamp_pb2

from vizier_amd._src.pyvizier import multimetric
from vizier_amd._src.pyvizier.study_config import StudyConfig
from vizier_amd._src.service import (
    constants,
    custom_errors,
    grpc_util,
    pythia_converters,
    resources,
)
from vizier_amd._src.service.proto import (
    operations_pb2,
    study_pb2,
    vizier_oss_pb2,
    vizier_service_pb2,
)
from vizier_amd._src.pythia import policy as pythia

logger = logging.getLogger(__name__)

_INTERNAL_CODE = 13  # google.rpc.Code.INTERNAL


def _now() -> timestamp_pb2.Timestamp:
  ts = timestamp_pb2.Timestamp()
  ts.GetCurrentTime()
  return ts


class VizierServicer:
  """Implements the VizierService RPCs (usable in-process or via gRPC)."""

  _TRIAL_MUTABLE_STATES = (
      study_pb2.Trial.State.Value('ACTIVE'),
      study_pb2.Trial.State.Value('STOPPING'),
  )

  def __init__(self, database_url: Optional[str] = constants.SQL_MEMORY_URL,
               early_stop_recycle_period: datetime.timedelta =
               datetime.timedelta(seconds=60),
               default_pythia_service=None):
    if database_url is None:
      from vizier_amd._src.service import ram_datastore
      self.datastore = ram_datastore.NestedDictRAMDataStore()
    else:
      import sqlalchemy as sqla
      from vizier_amd._src.service import sql_datastore
      engine = sqla.create_engine(
          database_url, connect_args={'check_same_thread': False},
          poolclass=sqla.pool.StaticPool)
      self.datastore = sql_datastore.SQLDataStore(engine)

    if default_pythia_service is None:
      from vizier_amd._src.service import pythia_service
      default_pythia_service = pythia_service.PythiaServicer(
          vizier_service=self)
    self.default_pythia_service = default_pythia_service

    self._owner_name_to_lock = collections.defaultdict(threading.Lock)
    self._study_name_to_lock = collections.defaultdict(threading.Lock)
    self._operation_lock = collections.defaultdict(threading.Lock)
    self._early_stop_recycle_period = early_stop_recycle_period

  # -- helpers --------------------------------------------------------------

  def _select_pythia_service(self, endpoint: Optional[str] = None):
    if endpoint is None or endpoint == constants.NO_ENDPOINT:
      return self.default_pythia_service
    from vizier_amd._src.service import stubs_util
    return stubs_util.create_pythia_server_stub(endpoint)

  def _study_is_immutable(self, study_name: str) -> bool:
    study = self.datastore.load_study(study_name)
    return study.state not in (study_pb2.Study.State.Value('ACTIVE'),
                               study_pb2.Study.State.Value(
                                   'STATE_UNSPECIFIED'))

  # -- study CRUD -----------------------------------------------------------

  def CreateStudy(self, request, context=None):
    study = request.study
    owner_id = resources.OwnerResource.from_name(request.parent).owner_id
    if study.name:
      grpc_util.handle_exception(
          ValueError('CreateStudy requests must not preset study.name.'),
          context)
    if not study.display_name:
      grpc_util.handle_exception(
          ValueError('Study display_name must be specified.'), context)

    with self._owner_name_to_lock[request.parent]:
      try:
        candidates = self.datastore.list_studies(request.parent)
      except custom_errors.NotFoundError:
        candidates = []
      if len(candidates) >= constants.MAX_STUDY_ID:
        grpc_util.handle_exception(
            ValueError(f'Maximum number of studies reached for {owner_id}.'),
            context)
      for candidate in candidates:
        if candidate.display_name == study.display_name:
          return candidate  # Idempotent load-on-exists.
      study.name = resources.StudyResource(owner_id,
                                           study.display_name).name
      self.datastore.create_study(study)
    return study

  def GetStudy(self, request, context=None):
    try:
      return self.datastore.load_study(request.name)
    except custom_errors.NotFoundError as e:
      grpc_util.handle_exception(e, context)

  def ListStudies(self, request, context=None):
    try:
      studies = self.datastore.list_studies(request.parent)
    except custom_errors.NotFoundError:
      studies = []
    return vizier_service_pb2.ListStudiesResponse(studies=studies)

  def DeleteStudy(self, request, context=None):
    try:
      self.datastore.delete_study(request.name)
    except custom_errors.NotFoundError as e:
      grpc_util.handle_exception(e, context)
    return empty_pb2.Empty()

  def SetStudyState(self, request, context=None):
    with self._study_name_to_lock[request.parent]:
      study = self.datastore.load_study(request.parent)
      study.state = request.state
      self.datastore.update_study(study)
    return study

  # -- suggestions ----------------------------------------------------------

  def SuggestTrials(self, request, context=None):
    study_name = request.parent
    if self._study_is_immutable(study_name):
      grpc_util.handle_exception(custom_errors.ImmutableStudyError(
          f'Study {study_name} is immutable. Cannot suggest trial.'), context)

    study_resource = resources.StudyResource.from_name(study_name)

    with self._operation_lock[study_name]:
      study = self.datastore.load_study(study_name)

      try:
        active_ops = self.datastore.list_suggestion_operations(
            study_name, request.client_id, lambda op: not op.done)
      except custom_errors.NotFoundError:
        active_ops = []
      if active_ops:
        return active_ops[0]

      start_time = _now()
      try:
        old_number = self.datastore.max_suggestion_operation_number(
            study_name, request.client_id)
      except custom_errors.NotFoundError:
        old_number = 0
      op_name = resources.SuggestionOperationResource(
          study_resource.owner_id, study_resource.study_id,
          request.client_id, old_number + 1).name
      output_op = operations_pb2.Operation(name=op_name, done=False)
      self.datastore.create_suggestion_operation(output_op)

      def _finish_with_trials(trials):
        resp = vizier_service_pb2.SuggestTrialsResponse(trials=trials)
        resp.start_time.CopyFrom(start_time)
        output_op.response.Pack(resp)
        output_op.done = True
        self.datastore.update_suggestion_operation(output_op)
        return output_op

      def _finish_with_error(message):
        output_op.error.code = _INTERNAL_CODE
        output_op.error.message = message
        output_op.done = True
        self.datastore.update_suggestion_operation(output_op)
        return output_op

      all_trials = self.datastore.list_trials(study_name)
      active_trials = [
          t for t in all_trials
          if t.state == study_pb2.Trial.State.Value('ACTIVE') and
          t.client_id == request.client_id]
      if len(active_trials) >= request.suggestion_count:
        return _finish_with_trials(active_trials[:request.suggestion_count])

      output_trials = active_trials
      requested = [t for t in all_trials
                   if t.state == study_pb2.Trial.State.Value('REQUESTED')]
      while requested and request.suggestion_count > len(output_trials):
        assigned = requested.pop()
        assigned.state = study_pb2.Trial.State.Value('ACTIVE')
        assigned.client_id = request.client_id
        assigned.start_time.CopyFrom(start_time)
        self.datastore.update_trial(assigned)
        output_trials.append(assigned)
      if len(output_trials) == request.suggestion_count:
        return _finish_with_trials(output_trials)

      # Ask Pythia for the missing suggestions.
      study_config = StudyConfig.from_proto(study.study_spec)
      descriptor = pythia.StudyDescriptor(
          config=study_config, guid=study_name,
          max_trial_id=self.datastore.max_trial_id(study_name))
      suggest_request = pythia.SuggestRequest(
          study_descriptor=descriptor,
          count=request.suggestion_count - len(output_trials))
      request_proto = pythia_converters.SuggestConverter.to_request_proto(
          suggest_request)
      request_proto.algorithm = study.study_spec.algorithm

      try:
        pythia_service = self._select_pythia_service(
            study_config.pythia_endpoint)
        decision_proto = pythia_service.Suggest(request_proto)
      except grpc.RpcError as e:
        logger.exception('Pythia failed to suggest trials')
        # Prefer gRPC details (carries the exception type for in-process
        # LocalRpcError, whose str() may be empty).
        details = e.details() if callable(getattr(e, 'details', None)) \
            else None
        return _finish_with_error(details or str(e) or type(e).__name__)

      decision = pythia_converters.SuggestConverter.from_decision_proto(
          decision_proto)

      try:
        from vizier_amd._src.pyvizier import metadata_util
        self.datastore.update_metadata(
            study_name,
            metadata_util.to_key_value_protos(decision.metadata.on_study),
            metadata_util.trial_metadata_to_update_list(
                decision.metadata.on_trials))
      except KeyError as e:
        logger.exception('Failed to write metadata update')
        return _finish_with_error(str(e))

      from vizier_amd._src.pyvizier import proto_converters as pc
      new_trials = pc.TrialConverter.to_protos(
          [s.to_trial() for s in decision.suggestions])

      if len(new_trials) < request.suggestion_count - len(output_trials):
        return _finish_with_error(
            f'Pythia under-delivered: needed '
            f'{request.suggestion_count - len(output_trials)}, got '
            f'{len(new_trials)}.')

      while request.suggestion_count > len(output_trials):
        new_trial = new_trials.pop()
        trial_id = self.datastore.max_trial_id(study_name) + 1
        new_trial.id = str(trial_id)
        new_trial.name = study_resource.trial_resource(trial_id).name
        new_trial.state = study_pb2.Trial.State.Value('ACTIVE')
        new_trial.start_time.CopyFrom(start_time)
        new_trial.client_id = request.client_id
        self.datastore.create_trial(new_trial)
        output_trials.append(new_trial)

      resp_op = _finish_with_trials(output_trials)

      # Bank any over-delivery into the REQUESTED pool.
      for remain in new_trials:
        trial_id = self.datastore.max_trial_id(study_name) + 1
        remain.id = str(trial_id)
        remain.name = study_resource.trial_resource(trial_id).name
        remain.state = study_pb2.Trial.State.Value('REQUESTED')
        self.datastore.create_trial(remain)

      return resp_op

  def GetOperation(self, request, context=None):
    try:
      return self.datastore.get_suggestion_operation(request.name)
    except custom_errors.NotFoundError as e:
      grpc_util.handle_exception(e, context)

  # -- trial CRUD -----------------------------------------------------------

  def CreateTrial(self, request, context=None):
    if self._study_is_immutable(request.parent):
      grpc_util.handle_exception(custom_errors.ImmutableStudyError(
          f'Study {request.parent} is immutable. Cannot create trial.'),
          context)
    trial = request.trial
    with self._study_name_to_lock[request.parent]:
      trial_id = self.datastore.max_trial_id(request.parent) + 1
      trial.id = str(trial_id)
      study_resource = resources.StudyResource.from_name(request.parent)
      trial.name = study_resource.trial_resource(trial_id).name
      if trial.state != study_pb2.Trial.State.Value('SUCCEEDED'):
        trial.state = study_pb2.Trial.State.Value('REQUESTED')
      trial.ClearField('client_id')
      trial.start_time.CopyFrom(_now())
      self.datastore.create_trial(trial)
    return trial

  def GetTrial(self, request, context=None):
    try:
      return self.datastore.get_trial(request.name)
    except custom_errors.NotFoundError as e:
      grpc_util.handle_exception(e, context)

  def ListTrials(self, request, context=None):
    try:
      trials = self.datastore.list_trials(request.parent)
    except custom_errors.NotFoundError as e:
      grpc_util.handle_exception(e, context)
    return vizier_service_pb2.ListTrialsResponse(trials=trials)

  def AddTrialMeasurement(self, request, context=None):
    study_name = resources.TrialResource.from_name(
        request.trial_name).study_resource.name
    if self._study_is_immutable(study_name):
      grpc_util.handle_exception(custom_errors.ImmutableStudyError(
          f'Study {study_name} is immutable. Cannot add measurement.'),
          context)
    with self._study_name_to_lock[study_name]:
      trial = self.datastore.get_trial(request.trial_name)
      if trial.state == study_pb2.Trial.State.Value('INFEASIBLE'):
        return trial
      if trial.state not in self._TRIAL_MUTABLE_STATES:
        grpc_util.handle_exception(custom_errors.ImmutableTrialError(
            f'Trial {request.trial_name} has state '
            f'{study_pb2.Trial.State.Name(trial.state)}; measurements can '
            'only be added in state ACTIVE or STOPPING.'), context)
      trial.measurements.add().CopyFrom(request.measurement)
      self.datastore.update_trial(trial)
    return trial

  def CompleteTrial(self, request, context=None):
    study_name = resources.TrialResource.from_name(
        request.name).study_resource.name
    if self._study_is_immutable(study_name):
      grpc_util.handle_exception(custom_errors.ImmutableStudyError(
          f'Study {study_name} is immutable. Cannot complete trial.'),
          context)
    with self._study_name_to_lock[study_name]:
      trial = self.datastore.get_trial(request.name)
      if trial.state not in self._TRIAL_MUTABLE_STATES:
        grpc_util.handle_exception(custom_errors.ImmutableTrialError(
            f'Trial {request.name} has state '
            f'{study_pb2.Trial.State.Name(trial.state)}; only ACTIVE or '
            'STOPPING trials can be completed.'), context)
      trial.state = study_pb2.Trial.State.Value('SUCCEEDED')
      if request.final_measurement.metrics:
        trial.final_measurement.CopyFrom(request.final_measurement)
      elif not request.trial_infeasible:
        if not trial.measurements:
          grpc_util.handle_exception(ValueError(
              'Both the request and trial intermediate measurements are '
              "missing. Cannot determine trial's final_measurement."),
              context)
        trial.final_measurement.CopyFrom(trial.measurements[-1])
      if request.trial_infeasible:
        trial.state = study_pb2.Trial.State.Value('INFEASIBLE')
        trial.infeasible_reason = request.infeasible_reason
      trial.end_time.CopyFrom(_now())
      self.datastore.update_trial(trial)
    return trial

  def DeleteTrial(self, request, context=None):
    study_name = resources.TrialResource.from_name(
        request.name).study_resource.name
    if self._study_is_immutable(study_name):
      grpc_util.handle_exception(custom_errors.ImmutableStudyError(
          f'Study {study_name} is immutable. Cannot delete trial.'), context)
    try:
      self.datastore.delete_trial(request.name)
    except custom_errors.NotFoundError as e:
      grpc_util.handle_exception(e, context)
    return empty_pb2.Empty()

  # -- early stopping -------------------------------------------------------

  def CheckTrialEarlyStoppingState(self, request, context=None):
    trial_resource = resources.TrialResource.from_name(request.trial_name)
    study_name = trial_resource.study_resource.name
    if self._study_is_immutable(study_name):
      grpc_util.handle_exception(custom_errors.ImmutableStudyError(
          f'Study {study_name} is immutable. Cannot early stop trial.'),
          context)
    with self._study_name_to_lock[study_name]:
      trial = self.datastore.get_trial(request.trial_name)
      if trial.state not in self._TRIAL_MUTABLE_STATES:
        grpc_util.handle_exception(custom_errors.ImmutableTrialError(
            f'Trial {request.trial_name} has state '
            f'{study_pb2.Trial.State.Name(trial.state)}; early stopping '
            'requires ACTIVE or STOPPING.'), context)

    op_name = trial_resource.early_stopping_operation_resource.name
    ACTIVE = vizier_oss_pb2.EarlyStoppingOperation.Status.Value('ACTIVE')
    DONE = vizier_oss_pb2.EarlyStoppingOperation.Status.Value('DONE')

    with self._operation_lock[study_name]:
      try:
        output_op = self.datastore.get_early_stopping_operation(op_name)
      except (custom_errors.NotFoundError, KeyError):
        output_op = None

      if output_op is None:
        output_op = vizier_oss_pb2.EarlyStoppingOperation(
            name=op_name, status=ACTIVE, should_stop=False)
        output_op.creation_time.CopyFrom(_now())
        self.datastore.create_early_stopping_operation(output_op)
      else:
        completion = output_op.completion_time.ToDatetime()
        age = datetime.datetime.now(datetime.timezone.utc).replace(
            tzinfo=None) - completion
        if (output_op.status == ACTIVE or
            age < self._early_stop_recycle_period):
          return vizier_service_pb2.CheckTrialEarlyStoppingStateResponse(
              should_stop=output_op.should_stop)
        output_op.status = ACTIVE
        output_op.should_stop = False
        self.datastore.update_early_stopping_operation(output_op)

      study = self.datastore.load_study(study_name)
      study_config = StudyConfig.from_proto(study.study_spec)
      descriptor = pythia.StudyDescriptor(
          config=study_config, guid=study_name,
          max_trial_id=self.datastore.max_trial_id(study_name))
      early_stop_request = pythia.EarlyStopRequest(
          study_descriptor=descriptor,
          trial_ids=[trial_resource.trial_id])
      request_proto = pythia_converters.EarlyStopConverter.to_request_proto(
          early_stop_request)
      spec_name = (study.study_spec.WhichOneof('automated_stopping_spec')
                   or 'default_stopping_spec')
      if spec_name != 'default_stopping_spec':
        raise ValueError(
            f'Misconfigured automated_stopping_spec: {study.study_spec}')
      request_proto.algorithm = 'RANDOM_SEARCH'

      pythia_service = self._select_pythia_service(
          study_config.pythia_endpoint)
      decisions_proto = pythia_service.EarlyStop(request_proto)
      decisions = pythia_converters.EarlyStopConverter.from_decisions_proto(
          decisions_proto)

      from vizier_amd._src.pyvizier import metadata_util
      self.datastore.update_metadata(
          study_name,
          metadata_util.to_key_value_protos(decisions.metadata.on_study),
          metadata_util.trial_metadata_to_update_list(
              decisions.metadata.on_trials))

      for decision in decisions.decisions:
        inner_name = resources.EarlyStoppingOperationResource(
            trial_resource.owner_id, trial_resource.study_id,
            decision.id).name
        try:
          inner_op = self.datastore.get_early_stopping_operation(inner_name)
        except (custom_errors.NotFoundError, KeyError):
          inner_op = vizier_oss_pb2.EarlyStoppingOperation(
              name=inner_name, status=ACTIVE, should_stop=False)
          inner_op.creation_time.CopyFrom(_now())
          self.datastore.create_early_stopping_operation(inner_op)
        inner_op.should_stop = decision.should_stop
        inner_op.status = DONE
        inner_op.completion_time.CopyFrom(_now())
        self.datastore.update_early_stopping_operation(inner_op)

      output_op = self.datastore.get_early_stopping_operation(op_name)
      return vizier_service_pb2.CheckTrialEarlyStoppingStateResponse(
          should_stop=output_op.should_stop)

  def StopTrial(self, request, context=None):
    study_name = resources.TrialResource.from_name(
        request.name).study_resource.name
    if self._study_is_immutable(study_name):
      grpc_util.handle_exception(custom_errors.ImmutableStudyError(
          f'Study {study_name} is immutable. Cannot stop trial.'), context)
    State = study_pb2.Trial.State
    with self._study_name_to_lock[study_name]:
      trial = self.datastore.get_trial(request.name)
      if trial.state == State.Value('ACTIVE'):
        trial.state = State.Value('STOPPING')
        self.datastore.update_trial(trial)
      elif trial.state in (State.Value('STOPPING'), State.Value('SUCCEEDED')):
        logger.warning('Trial %s already %s; StopTrial is a no-op.',
                       request.name, State.Name(trial.state))
      else:
        grpc_util.handle_exception(custom_errors.ImmutableTrialError(
            f'Trial {request.name} has state {State.Name(trial.state)}.'),
            context)
    return trial

  # -- analytics ------------------------------------------------------------

  def ListOptimalTrials(self, request, context=None):
    raw_trials = self.datastore.list_trials(request.parent)
    if not raw_trials:
      return vizier_service_pb2.ListOptimalTrialsResponse(optimal_trials=[])

    study_spec = self.datastore.load_study(request.parent).study_spec
    goals = {m.metric_id: m.goal for m in study_spec.metrics}
    required = set(goals)
    MINIMIZE = study_pb2.StudySpec.MetricSpec.GoalType.Value('MINIMIZE')

    considered, vectors = [], []
    for trial in raw_trials:
      values = {m.metric_id: m.value
                for m in trial.final_measurement.metrics}
      if (trial.state == study_pb2.Trial.State.Value('SUCCEEDED') and
          required.issubset(values)):
        vec = [(-values[mid] if goal == MINIMIZE else values[mid])
               for mid, goal in goals.items()]
        considered.append(trial)
        vectors.append(vec)
    if not considered:
      return vizier_service_pb2.ListOptimalTrialsResponse(optimal_trials=[])

    optimal = multimetric.is_pareto_optimal(np.asarray(vectors))
    return vizier_service_pb2.ListOptimalTrialsResponse(
        optimal_trials=[t for t, o in zip(considered, optimal) if o])

  def UpdateMetadata(self, request, context=None):
    if self._study_is_immutable(request.name):
      grpc_util.handle_exception(custom_errors.ImmutableStudyError(
          f'Study {request.name} is immutable. Cannot update metadata.'),
          context)
    try:
      self.datastore.update_metadata(
          request.name,
          [u.metadatum for u in request.delta if not u.HasField('trial_id')],
          [u for u in request.delta if u.HasField('trial_id')])
    except KeyError as e:
      return vizier_service_pb2.UpdateMetadataResponse(
          error_details=';'.join(str(a) for a in e.args))
    return vizier_service_pb2.UpdateMetadataResponse()
