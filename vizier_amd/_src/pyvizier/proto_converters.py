"""Bidirectional converters between pyvizier objects and the wire protos.

Capability parity with vizier/_src/pyvizier/oss/proto_converters.py
(ParameterConfigConverter :114, MeasurementConverter :391,
TrialConverter :577, SuggestConverter :869, ProblemStatementConverter :802).
"""

from __future__ import annotations

import datetime
from typing import Dict, Iterable, List, Optional, Sequence, Tuple, Union

from vizier_amd._src.pyvizier import metadata_util
from vizier_amd._src.pyvizier.base_study_config import (
    MetricInformation,
    MetricsConfig,
    ObjectiveMetricGoal,
    ProblemStatement,
)
from vizier_amd._src.pyvizier.common import Metadata, Namespace
from vizier_amd._src.pyvizier.parameter_config import (
    ExternalType,
    ParameterConfig,
    ParameterType,
    ScaleType,
    SearchSpace,
)
from vizier_amd._src.pyvizier.trial import (
    Measurement,
    Metric,
    ParameterDict,
    ParameterValue,
    Trial,
    TrialStatus,
    TrialSuggestion,
)
from vizier_amd._src.service.proto import pythia_service_pb2
from vizier_amd._src.service.proto import study_pb2

_SCALE_TO_PROTO = {
    ScaleType.LINEAR: study_pb2.StudySpec.ParameterSpec.ScaleType.Value(
        'UNIT_LINEAR_SCALE'),
    ScaleType.LOG: study_pb2.StudySpec.ParameterSpec.ScaleType.Value(
        'UNIT_LOG_SCALE'),
    ScaleType.REVERSE_LOG: study_pb2.StudySpec.ParameterSpec.ScaleType.Value(
        'UNIT_REVERSE_LOG_SCALE'),
}
_SCALE_FROM_PROTO = {v: k for k, v in _SCALE_TO_PROTO.items()}

_EXTERNAL_TO_PROTO = {
    ExternalType.INTERNAL: 0,
    ExternalType.BOOLEAN: 1,
    ExternalType.INTEGER: 2,
    ExternalType.FLOAT: 3,
}
_EXTERNAL_FROM_PROTO = {v: k for k, v in _EXTERNAL_TO_PROTO.items()}


def _timestamp_to_datetime(ts) -> datetime.datetime:
  return datetime.datetime.fromtimestamp(ts.seconds + 1e-9 * ts.nanos)


def _datetime_to_timestamp(dt: datetime.datetime, ts) -> None:
  secs = dt.timestamp()
  ts.seconds = int(secs)
  ts.nanos = round(1e9 * (secs - int(secs)))


class ParameterConfigConverter:
  """ParameterConfig <-> StudySpec.ParameterSpec."""

  @classmethod
  def to_proto(cls, pc: ParameterConfig) -> 'study_pb2.StudySpec.ParameterSpec':
    proto = study_pb2.StudySpec.ParameterSpec(parameter_id=pc.name)
    if pc.type == ParameterType.DOUBLE:
      lo, hi = pc.bounds
      proto.double_value_spec.min_value = lo
      proto.double_value_spec.max_value = hi
      if pc.default_value is not None:
        proto.double_value_spec.default_value.value = pc.default_value
    elif pc.type == ParameterType.INTEGER:
      lo, hi = pc.bounds
      proto.integer_value_spec.min_value = int(lo)
      proto.integer_value_spec.max_value = int(hi)
      if pc.default_value is not None:
        proto.integer_value_spec.default_value.value = int(pc.default_value)
    elif pc.type == ParameterType.CATEGORICAL:
      proto.categorical_value_spec.values.extend(pc.feasible_values)
      if pc.default_value is not None:
        proto.categorical_value_spec.default_value.value = pc.default_value
    elif pc.type == ParameterType.DISCRETE:
      proto.discrete_value_spec.values.extend(pc.feasible_values)
      if pc.default_value is not None:
        proto.discrete_value_spec.default_value.value = pc.default_value
    else:
      raise ValueError(f'Cannot convert parameter type {pc.type}')

    if pc.scale_type is not None:
      if pc.scale_type == ScaleType.UNIFORM_DISCRETE:
        # Not representable on the wire; drop it like the reference does.
        pass
      else:
        proto.scale_type = _SCALE_TO_PROTO[pc.scale_type]
    proto.external_type = _EXTERNAL_TO_PROTO[pc.external_type]

    # Group child configs that are identical under multiple parent values into
    # one ConditionalParameterSpec with a multi-value condition.
    grouped: List[Tuple[ParameterConfig, List]] = []
    for parent_value, subspace in pc.subspaces_by_value.items():
      for child in subspace.parameters:
        for child_proto_cfg, values in grouped:
          if child_proto_cfg == child:
            values.append(parent_value)
            break
        else:
          grouped.append((child, [parent_value]))
    for child, parent_values in grouped:
      cond = proto.conditional_parameter_specs.add()
      cond.parameter_spec.CopyFrom(cls.to_proto(child))
      if pc.type == ParameterType.DISCRETE:
        cond.parent_discrete_values.values.extend(
            float(v) for v in parent_values)
      elif pc.type == ParameterType.INTEGER:
        cond.parent_int_values.values.extend(int(v) for v in parent_values)
      elif pc.type == ParameterType.CATEGORICAL:
        cond.parent_categorical_values.values.extend(
            str(v) for v in parent_values)
      else:
        raise ValueError(
            f'{pc.type} parameters cannot have conditional children.')
    return proto

  @classmethod
  def from_proto(cls, proto,
                 *, strict_validation: bool = False) -> ParameterConfig:
    which = proto.WhichOneof('parameter_value_spec')
    default_value = None
    if which == 'double_value_spec':
      s = proto.double_value_spec
      bounds, feasible = (s.min_value, s.max_value), None
      if s.HasField('default_value'):
        default_value = s.default_value.value
    elif which == 'integer_value_spec':
      s = proto.integer_value_spec
      bounds, feasible = (int(s.min_value), int(s.max_value)), None
      if s.HasField('default_value'):
        default_value = int(s.default_value.value)
    elif which == 'categorical_value_spec':
      s = proto.categorical_value_spec
      bounds, feasible = None, list(s.values)
      if s.HasField('default_value'):
        default_value = s.default_value.value
    elif which == 'discrete_value_spec':
      s = proto.discrete_value_spec
      bounds, feasible = None, list(s.values)
      if s.HasField('default_value'):
        default_value = s.default_value.value
    else:
      raise ValueError(f'ParameterSpec has no value spec: {proto}')

    scale_type = _SCALE_FROM_PROTO.get(proto.scale_type, None)
    external_type = _EXTERNAL_FROM_PROTO.get(proto.external_type,
                                             ExternalType.INTERNAL)
    pc = ParameterConfig.factory(
        proto.parameter_id, bounds=bounds, feasible_values=feasible,
        scale_type=scale_type, default_value=default_value,
        external_type=external_type)

    for cond in proto.conditional_parameter_specs:
      child = cls.from_proto(cond.parameter_spec,
                             strict_validation=strict_validation)
      cond_which = cond.WhichOneof('parent_value_condition')
      if cond_which == 'parent_discrete_values':
        values = list(cond.parent_discrete_values.values)
      elif cond_which == 'parent_int_values':
        values = list(cond.parent_int_values.values)
      elif cond_which == 'parent_categorical_values':
        values = list(cond.parent_categorical_values.values)
      else:
        raise ValueError(f'Conditional spec without a condition: {cond}')
      for v in values:
        pc.add_child(v, child)
    return pc


class ParameterValueConverter:
  """ParameterValue <-> Trial.Parameter (google.protobuf.Value)."""

  @classmethod
  def to_proto(cls, value: ParameterValue, name: str
               ) -> 'study_pb2.Trial.Parameter':
    proto = study_pb2.Trial.Parameter(parameter_id=name)
    v = value.value
    if isinstance(v, bool):
      proto.value.bool_value = v
    elif isinstance(v, (int, float)):
      proto.value.number_value = float(v)
    elif isinstance(v, str):
      proto.value.string_value = v
    else:
      raise ValueError(f'Unsupported parameter value {v!r}')
    return proto

  @classmethod
  def from_proto(cls, proto) -> Optional[ParameterValue]:
    which = proto.value.WhichOneof('kind')
    if which == 'number_value':
      return ParameterValue(proto.value.number_value)
    if which == 'string_value':
      return ParameterValue(proto.value.string_value)
    if which == 'bool_value':
      return ParameterValue(proto.value.bool_value)
    return None


class MeasurementConverter:
  """Measurement <-> study_pb2.Measurement."""

  @classmethod
  def to_proto(cls, measurement: Measurement) -> 'study_pb2.Measurement':
    proto = study_pb2.Measurement()
    for name, metric in measurement.metrics.items():
      proto.metrics.add(metric_id=name, value=metric.value)
    proto.step_count = int(measurement.steps)
    secs = float(measurement.elapsed_secs)
    proto.elapsed_duration.seconds = int(secs)
    # round, not truncate: int() of e.g. 0.2s*1e9 = 199999999 nanos.
    proto.elapsed_duration.nanos = round(1e9 * (secs - int(secs)))
    return proto

  @classmethod
  def from_proto(cls, proto) -> Measurement:
    metrics: Dict[str, Metric] = {}
    for m in proto.metrics:
      # Last-write-wins on duplicated ids, like the reference.
      metrics[m.metric_id] = Metric(value=m.value)
    return Measurement(
        metrics=metrics,
        elapsed_secs=proto.elapsed_duration.seconds +
        1e-9 * proto.elapsed_duration.nanos,
        steps=proto.step_count)


def _trial_state_to_proto(status: TrialStatus, infeasible: bool) -> int:
  State = study_pb2.Trial.State
  if status == TrialStatus.COMPLETED:
    return State.Value('INFEASIBLE') if infeasible else State.Value(
        'SUCCEEDED')
  if status == TrialStatus.REQUESTED:
    return State.Value('REQUESTED')
  if status == TrialStatus.STOPPING:
    return State.Value('STOPPING')
  if status == TrialStatus.ACTIVE:
    return State.Value('ACTIVE')
  return State.Value('STATE_UNSPECIFIED')


class TrialConverter:
  """Trial <-> study_pb2.Trial."""

  @classmethod
  def from_proto(cls, proto) -> Trial:
    State = study_pb2.Trial.State
    parameters = {}
    for p in proto.parameters:
      value = ParameterValueConverter.from_proto(p)
      if value is None:
        continue
      if p.parameter_id in parameters:
        raise ValueError(f'Duplicate parameter {p.parameter_id} in {proto}')
      parameters[p.parameter_id] = value

    final_measurement = None
    if proto.HasField('final_measurement'):
      final_measurement = MeasurementConverter.from_proto(
          proto.final_measurement)

    completion_time = None
    infeasibility_reason = None
    if proto.state == State.Value('SUCCEEDED'):
      if proto.HasField('end_time'):
        completion_time = _timestamp_to_datetime(proto.end_time)
    elif proto.state == State.Value('INFEASIBLE'):
      infeasibility_reason = proto.infeasible_reason or ''

    metadata = metadata_util.from_key_value_protos(proto.metadata)
    measurements = [MeasurementConverter.from_proto(m)
                    for m in proto.measurements]

    creation_time = None
    if proto.HasField('start_time'):
      creation_time = _timestamp_to_datetime(proto.start_time)

    return Trial(
        id=int(proto.id) if proto.id else 0,
        description=proto.name,
        assigned_worker=proto.client_id or None,
        is_requested=proto.state == State.Value('REQUESTED'),
        stopping_reason=('stopping reason not supported yet'
                         if proto.state == State.Value('STOPPING') else None),
        parameters=parameters,
        creation_time=creation_time,
        completion_time=completion_time,
        infeasibility_reason=infeasibility_reason,
        final_measurement=final_measurement,
        measurements=measurements,
        metadata=metadata)

  @classmethod
  def from_protos(cls, protos: Iterable) -> List[Trial]:
    return [cls.from_proto(p) for p in protos]

  @classmethod
  def to_proto(cls, pytrial: Trial) -> 'study_pb2.Trial':
    proto = study_pb2.Trial()
    if pytrial.description is not None:
      proto.name = pytrial.description
    proto.id = str(pytrial.id)
    proto.state = _trial_state_to_proto(pytrial.status, pytrial.infeasible)
    proto.client_id = pytrial.assigned_worker or ''
    for name, value in pytrial.parameters.items():
      proto.parameters.append(ParameterValueConverter.to_proto(value, name))
    if pytrial.final_measurement is not None:
      proto.final_measurement.CopyFrom(
          MeasurementConverter.to_proto(pytrial.final_measurement))
    for m in pytrial.measurements:
      proto.measurements.append(MeasurementConverter.to_proto(m))
    if pytrial.creation_time is not None:
      _datetime_to_timestamp(pytrial.creation_time, proto.start_time)
    if pytrial.completion_time is not None:
      _datetime_to_timestamp(pytrial.completion_time, proto.end_time)
    if pytrial.infeasibility_reason is not None:
      proto.infeasible_reason = pytrial.infeasibility_reason
    for kv in metadata_util.to_key_value_protos(pytrial.metadata):
      proto.metadata.add().CopyFrom(kv)
    return proto

  @classmethod
  def to_protos(cls, pytrials: Iterable[Trial]) -> List:
    return [cls.to_proto(t) for t in pytrials]


class MetricInformationConverter:
  """MetricInformation <-> StudySpec.MetricSpec."""

  @classmethod
  def to_proto(cls, mi: MetricInformation) -> 'study_pb2.StudySpec.MetricSpec':
    proto = study_pb2.StudySpec.MetricSpec(
        metric_id=mi.name, goal=int(mi.goal))
    if mi.safety_threshold is not None:
      proto.safety_config.safety_threshold = mi.safety_threshold
      if mi.desired_min_safe_trials_fraction is not None:
        proto.safety_config.desired_min_safe_trials_fraction = (
            mi.desired_min_safe_trials_fraction)
    return proto

  @classmethod
  def from_proto(cls, proto) -> MetricInformation:
    goal = ObjectiveMetricGoal(proto.goal) if proto.goal else \
        ObjectiveMetricGoal.MAXIMIZE
    safety_threshold = None
    desired_fraction = None
    if proto.HasField('safety_config'):
      safety_threshold = proto.safety_config.safety_threshold
      if proto.safety_config.HasField('desired_min_safe_trials_fraction'):
        desired_fraction = (
            proto.safety_config.desired_min_safe_trials_fraction)
    return MetricInformation(
        name=proto.metric_id, goal=goal, safety_threshold=safety_threshold,
        desired_min_safe_trials_fraction=desired_fraction)


class SearchSpaceConverter:
  """SearchSpace <-> repeated StudySpec.ParameterSpec."""

  @classmethod
  def to_protos(cls, space: SearchSpace) -> List:
    return [ParameterConfigConverter.to_proto(p) for p in space.parameters]

  @classmethod
  def from_protos(cls, protos: Iterable) -> SearchSpace:
    space = SearchSpace()
    for p in protos:
      space.add(ParameterConfigConverter.from_proto(p))
    return space


class ProblemStatementConverter:
  """ProblemStatement <-> pythia_service_pb2.ProblemStatement."""

  @classmethod
  def to_proto(cls, problem: ProblemStatement
               ) -> 'pythia_service_pb2.ProblemStatement':
    proto = pythia_service_pb2.ProblemStatement()
    for p in SearchSpaceConverter.to_protos(problem.search_space):
      proto.search_space.add().CopyFrom(p)
    for mi in problem.metric_information:
      proto.metric_information.add().CopyFrom(
          MetricInformationConverter.to_proto(mi))
    for kv in metadata_util.to_key_value_protos(problem.metadata):
      proto.metadata.add().CopyFrom(kv)
    return proto

  @classmethod
  def from_proto(cls, proto) -> ProblemStatement:
    return ProblemStatement(
        search_space=SearchSpaceConverter.from_protos(proto.search_space),
        metric_information=MetricsConfig(
            MetricInformationConverter.from_proto(m)
            for m in proto.metric_information),
        metadata=metadata_util.from_key_value_protos(proto.metadata))


class TrialSuggestionConverter:
  """TrialSuggestion <-> pythia_service_pb2.TrialSuggestion."""

  @classmethod
  def to_proto(cls, suggestion: TrialSuggestion
               ) -> 'pythia_service_pb2.TrialSuggestion':
    proto = pythia_service_pb2.TrialSuggestion()
    for name, value in suggestion.parameters.items():
      proto.parameters.append(ParameterValueConverter.to_proto(value, name))
    for kv in metadata_util.to_key_value_protos(suggestion.metadata):
      proto.metadata.add().CopyFrom(kv)
    return proto

  @classmethod
  def from_proto(cls, proto) -> TrialSuggestion:
    params = {}
    for p in proto.parameters:
      value = ParameterValueConverter.from_proto(p)
      if value is not None:
        params[p.parameter_id] = value
    return TrialSuggestion(
        parameters=ParameterDict(params),
        metadata=metadata_util.from_key_value_protos(proto.metadata))


# Alias matching the reference's naming (proto_converters.py:869).
SuggestConverter = TrialSuggestionConverter


# Parameter sequence alias (oss/proto_converters.py:41).
from vizier_amd._src.pyvizier.parameter_config import (  # noqa: E402
    MonotypeParameterSequence,
)


class StudyStateConverter:
  """pythia StudyState <-> study_pb2.Study.State (oss converters :44)."""

  _TO_PROTO = {'ACTIVE': 'ACTIVE', 'ABORTED': 'INACTIVE',
               'COMPLETED': 'COMPLETED'}

  @classmethod
  def to_proto(cls, state):
    from vizier_amd._src.service.proto import study_pb2
    return study_pb2.Study.State.Value(cls._TO_PROTO[state.value])

  @classmethod
  def from_proto(cls, proto_state):
    from vizier_amd._src.pythia.policy import StudyState
    from vizier_amd._src.service.proto import study_pb2
    name = study_pb2.Study.State.Name(proto_state)
    for k, v in cls._TO_PROTO.items():
      if v == name:
        return StudyState(k)
    return StudyState.ACTIVE


class MetadataDeltaConverter:
  """MetadataDelta <-> UnitMetadataUpdate protos (oss converters :765)."""

  @classmethod
  def to_protos(cls, delta):
    from vizier_amd._src.pyvizier import metadata_util
    updates = metadata_util.study_metadata_to_update_list(delta.on_study)
    updates.extend(
        metadata_util.trial_metadata_to_update_list(delta.on_trials))
    return updates

