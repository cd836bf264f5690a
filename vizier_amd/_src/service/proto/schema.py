"""Runtime-built protobuf schema for the Vizier wire format.

The environment has no protoc/grpc_tools, so instead of checked-in
``*_pb2.py`` files we construct ``FileDescriptorProto``s programmatically
and register them in the default descriptor pool. Message/field names and
numbers mirror the reference protos exactly (the wire contract):

  * vizier/_src/service/key_value.proto
  * vizier/_src/service/study.proto
  * vizier/_src/service/vizier_oss.proto
  * vizier/_src/service/vizier_service.proto
  * vizier/_src/service/pythia_service.proto

plus wire-compatible ``google.longrunning.Operation`` / ``google.rpc.Status``
(absent from the installed protobuf wheel).
"""

from __future__ import annotations

from typing import Dict, List, Optional, Sequence, Tuple

from google.protobuf import descriptor_pb2
from google.protobuf import descriptor_pool
from google.protobuf import message_factory
# Importing the well-known types registers them in the default pool.
from google.protobuf import any_pb2  # noqa: F401
from google.protobuf import duration_pb2  # noqa: F401
from google.protobuf import empty_pb2  # noqa: F401
from google.protobuf import struct_pb2  # noqa: F401
from google.protobuf import timestamp_pb2  # noqa: F401
from google.protobuf import wrappers_pb2  # noqa: F401

_F = descriptor_pb2.FieldDescriptorProto

_TYPES = {
    'double': _F.TYPE_DOUBLE,
    'float': _F.TYPE_FLOAT,
    'int32': _F.TYPE_INT32,
    'int64': _F.TYPE_INT64,
    'bool': _F.TYPE_BOOL,
    'string': _F.TYPE_STRING,
    'bytes': _F.TYPE_BYTES,
}


def field(name: str, number: int, ftype: str, *, repeated: bool = False,
          oneof: Optional[int] = None, optional: bool = False) -> _F:
  """Builds a field. `ftype` is a scalar name, or '.pkg.Message'/'.pkg.Enum'."""
  f = _F()
  f.name = name
  f.number = number
  f.label = _F.LABEL_REPEATED if repeated else _F.LABEL_OPTIONAL
  if ftype in _TYPES:
    f.type = _TYPES[ftype]
  else:
    # Message vs enum resolved by the pool at build time; we mark message
    # unless the name is registered in _ENUM_NAMES below.
    f.type = _F.TYPE_ENUM if ftype in _ENUM_NAMES else _F.TYPE_MESSAGE
    f.type_name = ftype
  if oneof is not None:
    f.oneof_index = oneof
  if optional:
    # proto3 explicit presence: a synthetic oneof (must be added by caller).
    f.proto3_optional = True
  return f


def enum(name: str, values: Sequence[Tuple[str, int]]
         ) -> descriptor_pb2.EnumDescriptorProto:
  e = descriptor_pb2.EnumDescriptorProto()
  e.name = name
  for vname, vnum in values:
    v = e.value.add()
    v.name = vname
    v.number = vnum
  return e


def msg(name: str, fields: Sequence[_F] = (), *,
        oneofs: Sequence[str] = (),
        nested: Sequence[descriptor_pb2.DescriptorProto] = (),
        enums: Sequence[descriptor_pb2.EnumDescriptorProto] = (),
        reserved: Sequence[int] = ()) -> descriptor_pb2.DescriptorProto:
  m = descriptor_pb2.DescriptorProto()
  m.name = name
  oneof_count = len(oneofs)
  for o in oneofs:
    m.oneof_decl.add().name = o
  for f in fields:
    if f.proto3_optional and not f.HasField('oneof_index'):
      # Each proto3-optional field gets its own synthetic oneof.
      f.oneof_index = oneof_count
      m.oneof_decl.add().name = '_' + f.name
      oneof_count += 1
    m.field.add().CopyFrom(f)
  for n in nested:
    m.nested_type.add().CopyFrom(n)
  for e in enums:
    m.enum_type.add().CopyFrom(e)
  for r in reserved:
    rr = m.reserved_range.add()
    rr.start = r
    rr.end = r + 1
  return m


# Enum full names, so `field()` can mark TYPE_ENUM correctly.
_ENUM_NAMES = {
    '.vizier.Study.State',
    '.vizier.Trial.State',
    '.vizier.StudySpec.MetricSpec.GoalType',
    '.vizier.StudySpec.ParameterSpec.ScaleType',
    '.vizier.StudySpec.ParameterSpec.ExternalType',
    '.vizier.StudySpec.ObservationNoise',
    '.vizier.EarlyStoppingOperation.Status',
}


def _build_files() -> List[descriptor_pb2.FileDescriptorProto]:
  files: List[descriptor_pb2.FileDescriptorProto] = []

  def new_file(name: str, package: str,
               deps: Sequence[str] = ()) -> descriptor_pb2.FileDescriptorProto:
    fd = descriptor_pb2.FileDescriptorProto()
    fd.name = name
    fd.package = package
    fd.syntax = 'proto3'
    for d in deps:
      fd.dependency.append(d)
    files.append(fd)
    return fd

  # -- google/rpc/status.proto (wire-compatible stand-in) -------------------
  rpc = new_file('google/rpc/status.proto', 'google.rpc',
                 ['google/protobuf/any.proto'])
  rpc.message_type.add().CopyFrom(msg('Status', [
      field('code', 1, 'int32'),
      field('message', 2, 'string'),
      field('details', 3, '.google.protobuf.Any', repeated=True),
  ]))

  # -- google/longrunning/operations.proto (subset we use) ------------------
  lro = new_file('google/longrunning/operations.proto', 'google.longrunning',
                 ['google/protobuf/any.proto', 'google/rpc/status.proto'])
  lro.message_type.add().CopyFrom(msg('Operation', [
      field('name', 1, 'string'),
      field('metadata', 2, '.google.protobuf.Any'),
      field('done', 3, 'bool'),
      field('error', 4, '.google.rpc.Status', oneof=0),
      field('response', 5, '.google.protobuf.Any', oneof=0),
  ], oneofs=['result']))
  lro.message_type.add().CopyFrom(msg('GetOperationRequest', [
      field('name', 1, 'string'),
  ]))

  # -- key_value.proto ------------------------------------------------------
  kv = new_file('key_value.proto', 'vizier', ['google/protobuf/any.proto'])
  kv.message_type.add().CopyFrom(msg('KeyValue', [
      field('key', 1, 'string'),
      field('ns', 4, 'string'),
      field('value', 2, 'string', oneof=0),
      field('proto', 3, '.google.protobuf.Any', oneof=0),
  ], oneofs=['a_value']))

  # -- study.proto -----------------------------------------------------------
  st = new_file('study.proto', 'vizier', [
      'google/protobuf/duration.proto', 'google/protobuf/struct.proto',
      'google/protobuf/timestamp.proto', 'google/protobuf/wrappers.proto',
      'key_value.proto',
  ])
  st.message_type.add().CopyFrom(msg('Study', [
      field('name', 1, 'string'),
      field('display_name', 2, 'string'),
      field('study_spec', 3, '.vizier.StudySpec'),
      field('state', 4, '.vizier.Study.State'),
      field('create_time', 5, '.google.protobuf.Timestamp'),
      field('inactive_reason', 6, 'string'),
  ], enums=[enum('State', [('STATE_UNSPECIFIED', 0), ('ACTIVE', 1),
                           ('INACTIVE', 2), ('COMPLETED', 3)])]))

  st.message_type.add().CopyFrom(msg('Trial', [
      field('name', 1, 'string'),
      field('id', 2, 'string'),
      field('state', 3, '.vizier.Trial.State'),
      field('parameters', 4, '.vizier.Trial.Parameter', repeated=True),
      field('final_measurement', 5, '.vizier.Measurement'),
      field('measurements', 6, '.vizier.Measurement', repeated=True),
      field('start_time', 7, '.google.protobuf.Timestamp'),
      field('end_time', 8, '.google.protobuf.Timestamp'),
      field('client_id', 9, 'string'),
      field('infeasible_reason', 10, 'string'),
      field('metadata', 11, '.vizier.KeyValue', repeated=True),
  ], nested=[
      msg('Parameter', [
          field('parameter_id', 1, 'string'),
          field('value', 2, '.google.protobuf.Value'),
      ]),
  ], enums=[enum('State', [('STATE_UNSPECIFIED', 0), ('REQUESTED', 1),
                           ('ACTIVE', 2), ('STOPPING', 3), ('SUCCEEDED', 4),
                           ('INFEASIBLE', 5)])]))

  parameter_spec = msg('ParameterSpec', [
      field('parameter_id', 1, 'string'),
      field('double_value_spec', 2,
            '.vizier.StudySpec.ParameterSpec.DoubleValueSpec', oneof=0),
      field('integer_value_spec', 3,
            '.vizier.StudySpec.ParameterSpec.IntegerValueSpec', oneof=0),
      field('categorical_value_spec', 4,
            '.vizier.StudySpec.ParameterSpec.CategoricalValueSpec', oneof=0),
      field('discrete_value_spec', 5,
            '.vizier.StudySpec.ParameterSpec.DiscreteValueSpec', oneof=0),
      field('scale_type', 6, '.vizier.StudySpec.ParameterSpec.ScaleType'),
      field('external_type', 7,
            '.vizier.StudySpec.ParameterSpec.ExternalType'),
      field('conditional_parameter_specs', 10,
            '.vizier.StudySpec.ParameterSpec.ConditionalParameterSpec',
            repeated=True),
  ], oneofs=['parameter_value_spec'], nested=[
      msg('DoubleValueSpec', [
          field('min_value', 1, 'double'),
          field('max_value', 2, 'double'),
          field('default_value', 3, '.google.protobuf.DoubleValue'),
      ]),
      msg('IntegerValueSpec', [
          field('min_value', 1, 'int64'),
          field('max_value', 2, 'int64'),
          field('default_value', 3, '.google.protobuf.Int64Value'),
      ]),
      msg('CategoricalValueSpec', [
          field('values', 1, 'string', repeated=True),
          field('default_value', 2, '.google.protobuf.StringValue'),
      ]),
      msg('DiscreteValueSpec', [
          field('values', 1, 'double', repeated=True),
          field('default_value', 2, '.google.protobuf.DoubleValue'),
      ]),
      msg('ConditionalParameterSpec', [
          field('parameter_spec', 1, '.vizier.StudySpec.ParameterSpec'),
          field('parent_discrete_values', 2,
                '.vizier.StudySpec.ParameterSpec.ConditionalParameterSpec'
                '.DiscreteValueCondition', oneof=0),
          field('parent_int_values', 3,
                '.vizier.StudySpec.ParameterSpec.ConditionalParameterSpec'
                '.IntValueCondition', oneof=0),
          field('parent_categorical_values', 4,
                '.vizier.StudySpec.ParameterSpec.ConditionalParameterSpec'
                '.CategoricalValueCondition', oneof=0),
      ], oneofs=['parent_value_condition'], nested=[
          msg('DiscreteValueCondition',
              [field('values', 1, 'double', repeated=True)]),
          msg('IntValueCondition',
              [field('values', 1, 'int64', repeated=True)]),
          msg('CategoricalValueCondition',
              [field('values', 1, 'string', repeated=True)]),
      ]),
  ], enums=[
      enum('ScaleType', [('SCALE_TYPE_UNSPECIFIED', 0),
                         ('UNIT_LINEAR_SCALE', 1), ('UNIT_LOG_SCALE', 2),
                         ('UNIT_REVERSE_LOG_SCALE', 3)]),
      enum('ExternalType', [('AS_INTERNAL', 0), ('AS_BOOLEAN', 1),
                            ('AS_INTEGER', 2), ('AS_FLOAT', 3)]),
  ])

  metric_spec = msg('MetricSpec', [
      field('metric_id', 1, 'string'),
      field('goal', 2, '.vizier.StudySpec.MetricSpec.GoalType'),
      field('safety_config', 3,
            '.vizier.StudySpec.MetricSpec.SafetyMetricConfig'),
  ], nested=[
      msg('SafetyMetricConfig', [
          field('safety_threshold', 1, 'double'),
          field('desired_min_safe_trials_fraction', 2, 'double',
                optional=True),
      ]),
  ], enums=[enum('GoalType', [('GOAL_TYPE_UNSPECIFIED', 0), ('MAXIMIZE', 1),
                              ('MINIMIZE', 2)])])

  st.message_type.add().CopyFrom(msg('StudySpec', [
      field('metrics', 1, '.vizier.StudySpec.MetricSpec', repeated=True),
      field('parameters', 2, '.vizier.StudySpec.ParameterSpec', repeated=True),
      field('algorithm', 3, 'string'),
      field('default_stopping_spec', 9,
            '.vizier.StudySpec.DefaultEarlyStoppingSpec', oneof=0),
      field('observation_noise', 6, '.vizier.StudySpec.ObservationNoise'),
      field('metadata', 7, '.vizier.KeyValue', repeated=True),
  ], oneofs=['automated_stopping_spec'], nested=[
      metric_spec,
      parameter_spec,
      msg('DefaultEarlyStoppingSpec', []),
  ], enums=[
      enum('ObservationNoise', [('OBSERVATION_NOISE_UNSPECIFIED', 0),
                                ('LOW', 1), ('HIGH', 2)]),
  ], reserved=[4, 5, 8]))

  st.message_type.add().CopyFrom(msg('Measurement', [
      field('elapsed_duration', 1, '.google.protobuf.Duration'),
      field('step_count', 2, 'int64'),
      field('metrics', 3, '.vizier.Measurement.Metric', repeated=True),
  ], nested=[
      msg('Metric', [
          field('metric_id', 1, 'string'),
          field('value', 2, 'double'),
      ]),
  ]))

  # -- vizier_oss.proto ------------------------------------------------------
  oss = new_file('vizier_oss.proto', 'vizier',
                 ['google/protobuf/timestamp.proto'])
  oss.message_type.add().CopyFrom(msg('EarlyStoppingOperation', [
      field('name', 1, 'string'),
      field('status', 2, '.vizier.EarlyStoppingOperation.Status'),
      field('should_stop', 3, 'bool'),
      field('failure_message', 4, 'string'),
      field('creation_time', 5, '.google.protobuf.Timestamp'),
      field('completion_time', 6, '.google.protobuf.Timestamp'),
  ], enums=[enum('Status', [('UNKNOWN', 0), ('ACTIVE', 1), ('DONE', 2),
                            ('FAILED', 3)])]))

  # -- vizier_service.proto --------------------------------------------------
  vs = new_file('vizier_service.proto', 'vizier', [
      'google/longrunning/operations.proto', 'google/protobuf/empty.proto',
      'google/protobuf/timestamp.proto', 'key_value.proto', 'study.proto',
  ])
  M = vs.message_type
  M.add().CopyFrom(msg('GetStudyRequest', [field('name', 1, 'string')]))
  M.add().CopyFrom(msg('CreateStudyRequest', [
      field('parent', 1, 'string'),
      field('study', 2, '.vizier.Study'),
  ]))
  M.add().CopyFrom(msg('ListStudiesRequest', [
      field('parent', 1, 'string'),
      field('page_token', 2, 'string'),
      field('page_size', 3, 'int32'),
  ]))
  M.add().CopyFrom(msg('ListStudiesResponse', [
      field('studies', 1, '.vizier.Study', repeated=True),
      field('next_page_token', 2, 'string'),
  ]))
  M.add().CopyFrom(msg('DeleteStudyRequest', [field('name', 1, 'string')]))
  M.add().CopyFrom(msg('SetStudyStateRequest', [
      field('parent', 1, 'string'),
      field('state', 2, '.vizier.Study.State'),
  ]))
  M.add().CopyFrom(msg('SuggestTrialsRequest', [
      field('parent', 1, 'string'),
      field('suggestion_count', 2, 'int32'),
      field('client_id', 3, 'string'),
  ]))
  M.add().CopyFrom(msg('SuggestTrialsResponse', [
      field('trials', 1, '.vizier.Trial', repeated=True),
      field('study_state', 2, '.vizier.Study.State'),
      field('start_time', 3, '.google.protobuf.Timestamp'),
      field('end_time', 4, '.google.protobuf.Timestamp'),
  ]))
  M.add().CopyFrom(msg('CreateTrialRequest', [
      field('parent', 1, 'string'),
      field('trial', 2, '.vizier.Trial'),
  ]))
  M.add().CopyFrom(msg('GetTrialRequest', [field('name', 1, 'string')]))
  M.add().CopyFrom(msg('ListTrialsRequest', [
      field('parent', 1, 'string'),
      field('page_token', 2, 'string'),
      field('page_size', 3, 'int32'),
  ]))
  M.add().CopyFrom(msg('ListTrialsResponse', [
      field('trials', 1, '.vizier.Trial', repeated=True),
      field('next_page_token', 2, 'string'),
  ]))
  M.add().CopyFrom(msg('AddTrialMeasurementRequest', [
      field('trial_name', 1, 'string'),
      field('measurement', 3, '.vizier.Measurement'),
  ]))
  M.add().CopyFrom(msg('CompleteTrialRequest', [
      field('name', 1, 'string'),
      field('final_measurement', 2, '.vizier.Measurement'),
      field('trial_infeasible', 3, 'bool'),
      field('infeasible_reason', 4, 'string'),
  ]))
  M.add().CopyFrom(msg('DeleteTrialRequest', [field('name', 1, 'string')]))
  M.add().CopyFrom(msg('CheckTrialEarlyStoppingStateRequest', [
      field('trial_name', 1, 'string'),
  ]))
  M.add().CopyFrom(msg('CheckTrialEarlyStoppingStateResponse', [
      field('should_stop', 1, 'bool'),
  ]))
  M.add().CopyFrom(msg('StopTrialRequest', [field('name', 1, 'string')]))
  M.add().CopyFrom(msg('ListOptimalTrialsRequest', [
      field('parent', 1, 'string'),
      field('page_token', 2, 'string'),
      field('page_size', 3, 'int32'),
  ]))
  M.add().CopyFrom(msg('ListOptimalTrialsResponse', [
      field('optimal_trials', 1, '.vizier.Trial', repeated=True),
      field('next_page_token', 2, 'string'),
  ]))
  M.add().CopyFrom(msg('UnitMetadataUpdate', [
      field('trial_id', 3, 'string', optional=True),
      field('metadatum', 2, '.vizier.KeyValue'),
  ]))
  M.add().CopyFrom(msg('UpdateMetadataRequest', [
      field('name', 4, 'string'),
      field('delta', 2, '.vizier.UnitMetadataUpdate', repeated=True),
  ]))
  M.add().CopyFrom(msg('UpdateMetadataResponse', [
      field('error_details', 2, 'string'),
  ]))

  # -- pythia_service.proto --------------------------------------------------
  ps = new_file('pythia_service.proto', 'vizier', [
      'google/protobuf/empty.proto', 'key_value.proto', 'study.proto',
      'vizier_service.proto',
  ])
  P = ps.message_type
  P.add().CopyFrom(msg('TrialSuggestion', [
      field('parameters', 1, '.vizier.Trial.Parameter', repeated=True),
      field('metadata', 2, '.vizier.KeyValue', repeated=True),
  ]))
  P.add().CopyFrom(msg('ProblemStatement', [
      field('search_space', 1, '.vizier.StudySpec.ParameterSpec',
            repeated=True),
      field('metric_information', 2, '.vizier.StudySpec.MetricSpec',
            repeated=True),
      field('metadata', 3, '.vizier.KeyValue', repeated=True),
  ]))
  P.add().CopyFrom(msg('StudyDescriptor', [
      field('config', 1, '.vizier.ProblemStatement'),
      field('guid', 2, 'string'),
      field('max_trial_id', 3, 'int32'),
  ]))
  P.add().CopyFrom(msg('SuggestRequest', [
      field('algorithm', 1, 'string'),
      field('study_descriptor', 2, '.vizier.StudyDescriptor'),
      field('count', 3, 'int32'),
      field('checkpoint_dir', 4, 'string'),
  ]))
  P.add().CopyFrom(msg('SuggestDecision', [
      field('suggestions', 1, '.vizier.TrialSuggestion', repeated=True),
      field('metadata', 2, '.vizier.UnitMetadataUpdate', repeated=True),
  ]))
  P.add().CopyFrom(msg('EarlyStopRequest', [
      field('algorithm', 1, 'string'),
      field('study_descriptor', 2, '.vizier.StudyDescriptor'),
      field('trial_ids', 3, 'int32', repeated=True),
      field('checkpoint_dir', 4, 'string'),
  ]))
  P.add().CopyFrom(msg('EarlyStopDecision', [
      field('id', 1, 'int32'),
      field('reason', 2, 'string'),
      field('should_stop', 3, 'bool'),
      field('predicted_final_measurement', 4, '.vizier.Measurement',
            optional=True),
  ]))
  P.add().CopyFrom(msg('EarlyStopDecisions', [
      field('decisions', 1, '.vizier.EarlyStopDecision', repeated=True),
      field('metadata', 2, '.vizier.UnitMetadataUpdate', repeated=True),
  ]))

  return files


_POOL = descriptor_pool.Default()
_CLASSES: Dict[str, type] = {}


def _register() -> None:
  for fd in _build_files():
    try:
      _POOL.Add(fd)
    except Exception:
      # Already registered (module re-import); the existing file wins.
      pass


def get_message_class(full_name: str) -> type:
  cls = _CLASSES.get(full_name)
  if cls is None:
    cls = message_factory.GetMessageClass(
        _POOL.FindMessageTypeByName(full_name))
    _CLASSES[full_name] = cls
  return cls


_register()
