"""Conformance test mixin for StudyInterface implementations.

Capability parity with vizier/client/client_abc_testing.py:36-48: any
client implementation can subclass `TestCaseMixin`, provide
`create_study(problem, study_id)`, and inherit the conformance tests.
"""

from __future__ import annotations

import abc
from typing import Type

from vizier_amd import pyvizier as vz
from vizier_amd.client import client_abc


class TestCaseMixin(abc.ABC):
  """Mix into a unittest.TestCase/pytest class to test a client impl."""

  @abc.abstractmethod
  def create_study(self, problem: vz.ProblemStatement,
                   study_id: str) -> client_abc.StudyInterface:
    ...

  def _problem(self) -> vz.ProblemStatement:
    problem = vz.ProblemStatement()
    problem.search_space.root.add_float_param('x', 0.0, 1.0)
    problem.metric_information.append(vz.MetricInformation(
        name='obj', goal=vz.ObjectiveMetricGoal.MAXIMIZE))
    return problem

  def test_suggest_and_complete(self):
    study = self.create_study(self._problem(), 'conformance_suggest')
    trials = study.suggest(count=2)
    assert len(trials) == 2
    for t in trials:
      t.complete(vz.Measurement(metrics={'obj': 0.5}))
    materialized = [t for t in study.trials().get()]
    assert len(materialized) >= 2

  def test_request_and_get_trial(self):
    study = self.create_study(self._problem(), 'conformance_request')
    handle = study.request(vz.TrialSuggestion({'x': 0.25}))
    got = study.get_trial(handle.id)
    assert got.materialize().parameters.get_value('x') == 0.25

  def test_optimal_trials(self):
    study = self.create_study(self._problem(), 'conformance_optimal')
    for value in (0.1, 0.9, 0.5):
      for t in study.suggest(count=1):
        t.complete(vz.Measurement(metrics={'obj': value}))
    optimal = list(study.optimal_trials().get())
    assert optimal[0].final_measurement.metrics['obj'].value == 0.9

  def test_update_metadata(self):
    study = self.create_study(self._problem(), 'conformance_metadata')
    md = vz.Metadata()
    md['note'] = 'hello'
    study.update_metadata(md)
    config = study.materialize_problem_statement()
    assert config.metadata.get('note') == 'hello'
