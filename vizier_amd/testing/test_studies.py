"""Shared search-space fixtures for tests.

Capability parity with vizier/testing/test_studies.py:24-177.
"""

from __future__ import annotations

from vizier_amd import pyvizier as vz


def flat_continuous_space_with_scaling() -> vz.SearchSpace:
  space = vz.SearchSpace()
  root = space.root
  root.add_float_param('lineardouble', -1.0, 2.0)
  root.add_float_param('logdouble', 1e-4, 1e2,
                       scale_type=vz.ScaleType.LOG)
  return space


def flat_space_with_all_types() -> vz.SearchSpace:
  space = vz.SearchSpace()
  root = space.root
  root.add_float_param('lineardouble', -1.0, 2.0)
  root.add_float_param('logdouble', 1e-4, 1e2,
                       scale_type=vz.ScaleType.LOG)
  root.add_int_param('integer', -2, 2)
  root.add_categorical_param('categorical', ['a', 'aa', 'aaa'])
  root.add_bool_param('boolean')
  root.add_discrete_param('discrete_double', [-0.5, 1.0, 1.2])
  root.add_discrete_param('discrete_logdouble', [1e-5, 1e-2, 1e-1])
  root.add_discrete_param('discrete_int', [-1, 1, 2])
  return space


def conditional_automl_space() -> vz.SearchSpace:
  """'model_type' -> (learning_rate | optimizer) conditional tree."""
  space = vz.SearchSpace()
  root = space.root
  root.add_categorical_param('model_type', ['linear', 'dnn'])
  dnn = root.select('model_type', ['dnn'])
  dnn.add_float_param('learning_rate', 0.0001, 1.0,
                      default_value=0.001, scale_type=vz.ScaleType.LOG)
  linear = root.select('model_type', ['linear'])
  linear.add_float_param('learning_rate', 0.1, 1.0, default_value=0.1,
                         scale_type=vz.ScaleType.LOG)
  return space


def metrics_objective_goals() -> list:
  return [
      vz.MetricInformation(name='gain',
                           goal=vz.ObjectiveMetricGoal.MAXIMIZE),
      vz.MetricInformation(name='loss',
                           goal=vz.ObjectiveMetricGoal.MINIMIZE),
  ]


def metrics_all_unconstrained() -> list:
  return [
      vz.MetricInformation(name='max_nogoal',
                           goal=vz.ObjectiveMetricGoal.MAXIMIZE),
      vz.MetricInformation(name='min_nogoal',
                           goal=vz.ObjectiveMetricGoal.MINIMIZE),
  ]


def metrics_all_safe() -> list:
  return [
      vz.MetricInformation(name='safe_max',
                           goal=vz.ObjectiveMetricGoal.MAXIMIZE,
                           safety_threshold=-1.0),
      vz.MetricInformation(name='safe_min',
                           goal=vz.ObjectiveMetricGoal.MINIMIZE,
                           safety_threshold=1.0),
      vz.MetricInformation(name='objective',
                           goal=vz.ObjectiveMetricGoal.MAXIMIZE),
  ]
