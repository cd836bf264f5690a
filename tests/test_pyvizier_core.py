"""Tests for the pyvizier data model (namespaces, search space, trials)."""

import datetime

import numpy as np
import pytest

from vizier_amd import pyvizier as vz
from vizier_amd._src.pyvizier import multimetric
from vizier_amd._src.pyvizier.common import Namespace


class TestNamespace:

  def test_encode_decode_roundtrip(self):
    cases = [(), ('a',), ('a', 'b'), ('',), ('', ''), ('a:b',), ('a', 'b:c')]
    for parts in cases:
      ns = Namespace(parts)
      assert Namespace.decode(ns.encode()) == ns, parts

  def test_encoding_matches_reference_format(self):
    assert Namespace(()).encode() == ''
    assert Namespace(('a',)).encode() == ':a'
    assert Namespace(('a', 'b')).encode() == ':a:b'
    assert Namespace(('a:b',)).encode() == ':a\\:b'
    assert Namespace.decode('a:b') == Namespace(('a', 'b'))
    assert Namespace.decode(':a') == Namespace(('a',))
    assert Namespace.decode('a\\:b') == Namespace(('a:b',))

  def test_startswith(self):
    ns = Namespace(('a', 'b', 'c'))
    assert ns.startswith(())
    assert ns.startswith(('a',))
    assert ns.startswith(('a', 'b', 'c'))
    assert not ns.startswith(('b',))


class TestMetadata:

  def test_basic_mapping(self):
    md = vz.Metadata()
    md['k'] = 'v'
    assert md['k'] == 'v'
    assert len(md) == 1

  def test_namespace_views_share_store(self):
    md = vz.Metadata()
    md.ns('algo')['state'] = 'x'
    assert md.abs_ns(Namespace(('algo',)))['state'] == 'x'
    assert 'state' not in md
    assert Namespace(('algo',)) in md.namespaces()

  def test_proto_values(self):
    from google.protobuf import any_pb2
    from vizier_amd._src.service.proto import study_pb2
    md = vz.Metadata()
    m = study_pb2.Measurement()
    m.step_count = 7
    md['proto'] = m
    got = md.get_proto('proto', cls=study_pb2.Measurement)
    assert got.step_count == 7
    any_msg = any_pb2.Any()
    any_msg.Pack(m)
    md['any'] = any_msg
    got2 = md.get_proto('any', cls=study_pb2.Measurement)
    assert got2.step_count == 7

  def test_attach(self):
    a = vz.Metadata()
    a.ns('x')['k'] = '1'
    b = vz.Metadata()
    b.ns('y')['k'] = '2'
    a.attach(b)
    assert a.abs_ns(('y',))['k'] == '2'


class TestSearchSpace:

  def test_add_params_and_types(self):
    space = vz.SearchSpace()
    root = space.root
    root.add_float_param('lr', 1e-4, 1.0, scale_type=vz.ScaleType.LOG)
    root.add_int_param('units', 8, 128)
    root.add_discrete_param('batch', [16, 32, 64])
    root.add_categorical_param('opt', ['adam', 'sgd'], default_value='adam')
    root.add_bool_param('nesterov')
    assert space.parameter_names == ['lr', 'units', 'batch', 'opt',
                                     'nesterov']
    assert space.get('lr').type == vz.ParameterType.DOUBLE
    assert space.get('units').type == vz.ParameterType.INTEGER
    assert space.get('batch').type == vz.ParameterType.DISCRETE
    assert space.get('opt').type == vz.ParameterType.CATEGORICAL
    assert space.get('opt').default_value == 'adam'
    assert space.get('nesterov').feasible_values == ['false', 'true']
    assert space.get('batch').bounds == (16.0, 64.0)
    assert space.get('units').num_feasible_values == 121
    assert not space.is_conditional

  def test_duplicate_name_raises(self):
    space = vz.SearchSpace()
    space.root.add_float_param('x', 0, 1)
    with pytest.raises(ValueError):
      space.root.add_float_param('x', 0, 1)

  def test_conditional_space(self):
    space = vz.SearchSpace()
    root = space.root
    root.add_categorical_param('model', ['dnn', 'linear'])
    dnn = root.select('model', ['dnn'])
    dnn.add_int_param('hidden', 1, 10)
    assert space.is_conditional
    assert space.num_parameters() == 2
    children = space.get('model').child_parameter_configs
    assert [c.name for c in children] == ['hidden']

  def test_multi_subspace_select(self):
    space = vz.SearchSpace()
    root = space.root
    root.add_categorical_param('model', ['dnn', 'linear'])
    sel = root.select('model', ['dnn', 'linear'])
    sel.add_categorical_param('optimizer', ['adam', 'adagrad'])
    assert space.num_parameters() == 3  # model + 2 copies of optimizer

  def test_contains_and_cast(self):
    cfg = vz.ParameterConfig.factory('x', bounds=(0.0, 1.0))
    assert cfg.contains(0.5)
    assert not cfg.contains(1.5)
    icfg = vz.ParameterConfig.factory('i', bounds=(1, 5))
    assert icfg.cast_value(3.0) == 3
    assert icfg.contains(5) and not icfg.contains(6)
    dcfg = vz.ParameterConfig.factory('d', feasible_values=[1.0, 2.5])
    assert dcfg.contains(2.5) and not dcfg.contains(2.0)
    assert dcfg.round_to_feasible(2.1) == 2.5

  def test_default_value_validation(self):
    with pytest.raises(vz.InvalidParameterError):
      vz.ParameterConfig.factory('x', bounds=(0.0, 1.0), default_value=3.0)


class TestTrial:

  def test_parameter_value_casting(self):
    assert vz.ParameterValue('true').as_bool is True
    assert vz.ParameterValue('true').as_float == 1.0
    assert vz.ParameterValue(1).as_bool is True
    assert vz.ParameterValue(0.5).as_float == 0.5
    assert vz.ParameterValue('x').as_float is None
    assert vz.ParameterValue(True).as_str == 'true'

  def test_trial_lifecycle(self):
    t = vz.Trial(parameters={'x': 0.5}, id=3)
    assert t.status == vz.TrialStatus.ACTIVE
    assert not t.is_completed
    t.complete(vz.Measurement(metrics={'loss': 0.1}))
    assert t.status == vz.TrialStatus.COMPLETED
    assert t.is_completed
    assert t.final_measurement.metrics['loss'].value == 0.1
    assert t.duration is not None

  def test_infeasible(self):
    t = vz.Trial(id=1)
    t.complete(vz.Measurement(), infeasibility_reason='nan')
    assert t.infeasible
    assert t.status == vz.TrialStatus.COMPLETED
    assert t.infeasibility_reason == 'nan'

  def test_requested_and_stopping(self):
    t = vz.Trial(id=2, is_requested=True)
    assert t.status == vz.TrialStatus.REQUESTED
    t2 = vz.Trial(id=3, stopping_reason='stop')
    assert t2.status == vz.TrialStatus.STOPPING

  def test_suggestion_to_trial(self):
    s = vz.TrialSuggestion({'a': 1})
    t = s.to_trial(5)
    assert t.id == 5
    assert t.parameters.get_value('a') == 1

  def test_measurement_validation(self):
    with pytest.raises(ValueError):
      vz.Measurement(elapsed_secs=-1)
    m = vz.Measurement(metrics={'m': 2})
    assert m.metrics['m'] == vz.Metric(value=2.0)

  def test_trial_filter(self):
    trials = [vz.Trial(id=i) for i in range(5)]
    trials[0].complete(vz.Measurement())
    f = vz.TrialFilter(min_id=1, status=[vz.TrialStatus.ACTIVE])
    assert [t.id for t in trials if f(t)] == [1, 2, 3, 4]


class TestMetricInformation:

  def test_flip_goal(self):
    mi = vz.MetricInformation(name='m', goal=vz.ObjectiveMetricGoal.MAXIMIZE)
    assert mi.goal.is_maximize
    mi.flip_goal()
    assert mi.goal.is_minimize

  def test_safety_type(self):
    mi = vz.MetricInformation(name='s', goal=vz.ObjectiveMetricGoal.MAXIMIZE,
                              safety_threshold=0.5)
    assert mi.type == vz.MetricType.SAFETY
    config = vz.MetricsConfig([
        mi, vz.MetricInformation(name='o',
                                 goal=vz.ObjectiveMetricGoal.MINIMIZE)])
    assert config.is_single_objective
    assert config.is_safety_metric

  def test_duplicate_names(self):
    with pytest.raises(ValueError):
      vz.MetricsConfig([
          vz.MetricInformation(name='m', goal=1),
          vz.MetricInformation(name='m', goal=2)])


class TestMultimetric:

  def test_pareto_simple(self):
    pts = np.array([[1, 0], [0, 1], [0.5, 0.5], [0.2, 0.2]])
    opt = multimetric.is_pareto_optimal(pts)
    assert list(opt) == [True, True, True, False]

  def test_pareto_duplicates_optimal(self):
    pts = np.array([[1.0, 1.0], [1.0, 1.0], [0.0, 0.0]])
    opt = multimetric.is_pareto_optimal(pts)
    assert list(opt) == [True, True, False]

  def test_fast_matches_naive(self):
    rng = np.random.default_rng(0)
    pts = rng.standard_normal((500, 3))
    fast = multimetric.FastParetoOptimalAlgorithm(recursive_threshold=50)
    naive = multimetric.NaiveParetoOptimalAlgorithm()
    np.testing.assert_array_equal(fast.is_pareto_optimal(pts),
                                  naive.is_pareto_optimal(pts))

  def test_hypervolume_unit_square(self):
    # One point at (1,1) from origin: HV = 1.
    front = multimetric.ParetoFrontier(np.array([[1.0, 1.0]]),
                                       origin=np.zeros(2),
                                       num_vectors=20000, seed=1)
    hv = front.hypervolume()
    assert abs(float(hv) - 1.0) < 0.05

  def test_cumulative_hypervolume_monotone(self):
    rng = np.random.default_rng(2)
    pts = rng.uniform(0, 1, size=(20, 2))
    front = multimetric.ParetoFrontier(pts, origin=np.zeros(2),
                                       num_vectors=2000, seed=3)
    cum = front.hypervolume(is_cumulative=True)
    assert np.all(np.diff(cum) >= -1e-12)

  def test_safety_checker(self):
    cfg = vz.MetricsConfig([
        vz.MetricInformation(name='obj', goal=vz.ObjectiveMetricGoal.MAXIMIZE),
        vz.MetricInformation(name='safe', goal=vz.ObjectiveMetricGoal.MAXIMIZE,
                             safety_threshold=0.5)])
    checker = multimetric.SafetyChecker(cfg)
    ms = [vz.Measurement(metrics={'obj': 1, 'safe': 0.7}),
          vz.Measurement(metrics={'obj': 1, 'safe': 0.3}),
          vz.Measurement(metrics={'obj': 1})]
    assert checker.are_measurements_safe(ms) == [True, False, True]
