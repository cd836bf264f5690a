"""Round-trip tests for pyvizier <-> proto converters."""

import pytest

from vizier_amd import pyvizier as vz
from vizier_amd._src.pyvizier import proto_converters as pc
from vizier_amd._src.service.proto import study_pb2


def _mixed_space() -> vz.SearchSpace:
  space = vz.SearchSpace()
  root = space.root
  root.add_float_param('lr', 1e-4, 1.0, scale_type=vz.ScaleType.LOG,
                       default_value=0.01)
  root.add_int_param('units', 8, 128)
  root.add_discrete_param('batch', [16, 32, 64], default_value=32)
  root.add_categorical_param('opt', ['adam', 'sgd'], default_value='sgd')
  return space


class TestParameterConfigConverter:

  def test_roundtrip_types(self):
    for cfg in _mixed_space().parameters:
      proto = pc.ParameterConfigConverter.to_proto(cfg)
      back = pc.ParameterConfigConverter.from_proto(proto)
      assert back == cfg, cfg.name

  def test_scale_types_on_wire(self):
    cfg = vz.ParameterConfig.factory('lr', bounds=(1e-4, 1.0),
                                     scale_type=vz.ScaleType.LOG)
    proto = pc.ParameterConfigConverter.to_proto(cfg)
    assert proto.scale_type == \
        study_pb2.StudySpec.ParameterSpec.ScaleType.Value('UNIT_LOG_SCALE')

  def test_conditional_roundtrip(self):
    space = vz.SearchSpace()
    root = space.root
    root.add_categorical_param('model', ['dnn', 'linear'])
    root.select('model', ['dnn']).add_int_param('hidden', 1, 10)
    root.select('model', ['dnn', 'linear']).add_float_param('reg', 0.0, 1.0)
    cfg = space.get('model')
    proto = pc.ParameterConfigConverter.to_proto(cfg)
    # 'reg' appears under both parent values -> grouped into one conditional
    # spec with two condition values.
    assert len(proto.conditional_parameter_specs) == 2
    back = pc.ParameterConfigConverter.from_proto(proto)
    assert back == cfg


class TestTrialConverter:

  def test_roundtrip_completed(self):
    t = vz.Trial(parameters={'x': 0.5, 'c': 'red', 'b': True}, id=7)
    t.metadata.ns('algo')['state'] = 'blob'
    t.complete(vz.Measurement(metrics={'loss': 0.25}, elapsed_secs=2.5,
                              steps=10))
    proto = pc.TrialConverter.to_proto(t)
    assert proto.state == study_pb2.Trial.State.Value('SUCCEEDED')
    assert proto.id == '7'
    back = pc.TrialConverter.from_proto(proto)
    assert back.id == 7
    assert back.parameters.get_value('x') == 0.5
    assert back.parameters.get_value('c') == 'red'
    assert back.parameters.get_value('b') is True
    assert back.final_measurement.metrics['loss'].value == 0.25
    assert back.final_measurement.elapsed_secs == pytest.approx(2.5)
    assert back.metadata.abs_ns(('algo',))['state'] == 'blob'
    assert back.status == vz.TrialStatus.COMPLETED

  def test_infeasible(self):
    t = vz.Trial(id=1)
    t.complete(vz.Measurement(), infeasibility_reason='broke')
    proto = pc.TrialConverter.to_proto(t)
    assert proto.state == study_pb2.Trial.State.Value('INFEASIBLE')
    back = pc.TrialConverter.from_proto(proto)
    assert back.infeasible and back.infeasibility_reason == 'broke'

  def test_requested_state(self):
    t = vz.Trial(id=2, is_requested=True)
    proto = pc.TrialConverter.to_proto(t)
    assert proto.state == study_pb2.Trial.State.Value('REQUESTED')
    back = pc.TrialConverter.from_proto(proto)
    assert back.status == vz.TrialStatus.REQUESTED


class TestStudyConfig:

  def test_roundtrip(self):
    sc = vz.StudyConfig(
        search_space=_mixed_space(),
        metric_information=[vz.MetricInformation(
            name='loss', goal=vz.ObjectiveMetricGoal.MINIMIZE)],
        algorithm=vz.Algorithm.RANDOM_SEARCH,
        observation_noise=vz.ObservationNoise.HIGH)
    sc.metadata['note'] = 'hello'
    proto = sc.to_proto()
    assert proto.algorithm == 'RANDOM_SEARCH'
    back = vz.StudyConfig.from_proto(proto)
    assert back.algorithm == 'RANDOM_SEARCH'
    assert back.observation_noise == vz.ObservationNoise.HIGH
    assert back.search_space == sc.search_space
    assert back.metric_information == sc.metric_information
    assert back.metadata['note'] == 'hello'
    # Round-trip again: stable.
    assert vz.StudyConfig.from_proto(back.to_proto()).to_proto() == \
        back.to_proto()

  def test_automated_stopping(self):
    sc = vz.StudyConfig(
        search_space=_mixed_space(),
        metric_information=[vz.MetricInformation(
            name='m', goal=vz.ObjectiveMetricGoal.MAXIMIZE)],
        automated_stopping_config=(
            vz.AutomatedStoppingConfig.default_stopping_spec()))
    proto = sc.to_proto()
    assert proto.WhichOneof('automated_stopping_spec') == \
        'default_stopping_spec'
    back = vz.StudyConfig.from_proto(proto)
    assert back.automated_stopping_config is not None

  def test_pythia_endpoint_metadata(self):
    sc = vz.StudyConfig(search_space=_mixed_space(),
                        metric_information=[vz.MetricInformation(
                            name='m', goal=1)],
                        pythia_endpoint='localhost:1234')
    proto = sc.to_proto()
    back = vz.StudyConfig.from_proto(proto)
    assert back.pythia_endpoint == 'localhost:1234'

  def test_trial_parameters_external_types(self):
    space = vz.SearchSpace()
    space.root.add_bool_param('flag')
    space.root.add_discrete_param('d', [1, 2, 4])
    sc = vz.StudyConfig(search_space=space, metric_information=[
        vz.MetricInformation(name='m', goal=1)])
    t = vz.Trial(parameters={'flag': 'true', 'd': 2.0})
    vals = sc.pytrial_parameters(t)
    assert vals['flag'] is True
    assert vals['d'] == 2 and isinstance(vals['d'], int)


class TestProblemStatementConverter:

  def test_roundtrip(self):
    problem = vz.ProblemStatement(
        search_space=_mixed_space(),
        metric_information=[vz.MetricInformation(
            name='acc', goal=vz.ObjectiveMetricGoal.MAXIMIZE)])
    problem.metadata.ns('x')['k'] = 'v'
    proto = pc.ProblemStatementConverter.to_proto(problem)
    back = pc.ProblemStatementConverter.from_proto(proto)
    assert back == problem


class TestSuggestionConverter:

  def test_roundtrip(self):
    s = vz.TrialSuggestion({'x': 1.5, 'c': 'a'})
    s.metadata.ns('eagle')['fly'] = 'yes'
    proto = pc.TrialSuggestionConverter.to_proto(s)
    back = pc.TrialSuggestionConverter.from_proto(proto)
    assert back.parameters == s.parameters
    assert back.metadata.abs_ns(('eagle',))['fly'] == 'yes'
