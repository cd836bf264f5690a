"""Service tests: in-process servicer, gRPC loopback, suggest state machine."""

import threading

import pytest

from vizier_amd import pyvizier as vz
from vizier_amd._src.service import custom_errors, resources
from vizier_amd._src.service.proto import study_pb2, vizier_service_pb2
from vizier_amd._src.service.vizier_service import VizierServicer
from vizier_amd.service import (
    DefaultVizierServer,
    DistributedPythiaVizierServer,
    clients,
)


def make_study_config(algorithm='RANDOM_SEARCH') -> vz.StudyConfig:
  config = vz.StudyConfig(algorithm=algorithm)
  root = config.search_space.root
  root.add_float_param('x', 0.0, 1.0)
  root.add_int_param('i', 1, 10)
  root.add_categorical_param('c', ['a', 'b'])
  root.add_discrete_param('d', [1.0, 2.0, 4.0])
  config.metric_information.append(
      vz.MetricInformation(name='obj', goal=vz.ObjectiveMetricGoal.MAXIMIZE))
  return config


def create_study(servicer, owner='o', display_name='s',
                 algorithm='RANDOM_SEARCH'):
  study = study_pb2.Study(display_name=display_name)
  study.study_spec.CopyFrom(make_study_config(algorithm).to_proto())
  return servicer.CreateStudy(vizier_service_pb2.CreateStudyRequest(
      parent=f'owners/{owner}', study=study))


class TestVizierServicerInProcess:
  """Exercises the servicer without gRPC (context=None path)."""

  def setup_method(self):
    self.servicer = VizierServicer(database_url=None)

  def test_create_study_idempotent(self):
    s1 = create_study(self.servicer)
    s2 = create_study(self.servicer)
    assert s1.name == s2.name == 'owners/o/studies/s'

  def test_suggest_trials_creates_active_trials(self):
    study = create_study(self.servicer)
    op = self.servicer.SuggestTrials(vizier_service_pb2.SuggestTrialsRequest(
        parent=study.name, suggestion_count=3, client_id='worker0'))
    assert op.done
    resp = vizier_service_pb2.SuggestTrialsResponse.FromString(
        op.response.value)
    assert len(resp.trials) == 3
    for t in resp.trials:
      assert t.state == study_pb2.Trial.State.Value('ACTIVE')
      assert t.client_id == 'worker0'

  def test_suggest_reuses_active_trials_per_client(self):
    study = create_study(self.servicer)
    op1 = self.servicer.SuggestTrials(vizier_service_pb2.SuggestTrialsRequest(
        parent=study.name, suggestion_count=2, client_id='w'))
    r1 = vizier_service_pb2.SuggestTrialsResponse.FromString(
        op1.response.value)
    # Same client asks again without completing: same trials, no new ones.
    op2 = self.servicer.SuggestTrials(vizier_service_pb2.SuggestTrialsRequest(
        parent=study.name, suggestion_count=2, client_id='w'))
    r2 = vizier_service_pb2.SuggestTrialsResponse.FromString(
        op2.response.value)
    assert [t.id for t in r1.trials] == [t.id for t in r2.trials]
    # Different client gets different trials.
    op3 = self.servicer.SuggestTrials(vizier_service_pb2.SuggestTrialsRequest(
        parent=study.name, suggestion_count=1, client_id='w2'))
    r3 = vizier_service_pb2.SuggestTrialsResponse.FromString(
        op3.response.value)
    assert r3.trials[0].id not in [t.id for t in r1.trials]

  def test_requested_pool_assignment(self):
    study = create_study(self.servicer)
    # User-created trial goes into REQUESTED pool...
    req = vizier_service_pb2.CreateTrialRequest(parent=study.name)
    p = req.trial.parameters.add(parameter_id='x')
    p.value.number_value = 0.25
    created = self.servicer.CreateTrial(req)
    assert created.state == study_pb2.Trial.State.Value('REQUESTED')
    # ... and is drained by the next suggestion.
    op = self.servicer.SuggestTrials(vizier_service_pb2.SuggestTrialsRequest(
        parent=study.name, suggestion_count=1, client_id='w'))
    resp = vizier_service_pb2.SuggestTrialsResponse.FromString(
        op.response.value)
    assert resp.trials[0].id == created.id
    assert resp.trials[0].state == study_pb2.Trial.State.Value('ACTIVE')

  def test_complete_trial_auto_selects_final_measurement(self):
    study = create_study(self.servicer)
    op = self.servicer.SuggestTrials(vizier_service_pb2.SuggestTrialsRequest(
        parent=study.name, suggestion_count=1, client_id='w'))
    trial = vizier_service_pb2.SuggestTrialsResponse.FromString(
        op.response.value).trials[0]
    add = vizier_service_pb2.AddTrialMeasurementRequest(
        trial_name=trial.name)
    add.measurement.metrics.add(metric_id='obj', value=0.7)
    self.servicer.AddTrialMeasurement(add)
    done = self.servicer.CompleteTrial(
        vizier_service_pb2.CompleteTrialRequest(name=trial.name))
    assert done.state == study_pb2.Trial.State.Value('SUCCEEDED')
    assert done.final_measurement.metrics[0].value == 0.7

  def test_complete_infeasible(self):
    study = create_study(self.servicer)
    op = self.servicer.SuggestTrials(vizier_service_pb2.SuggestTrialsRequest(
        parent=study.name, suggestion_count=1, client_id='w'))
    trial = vizier_service_pb2.SuggestTrialsResponse.FromString(
        op.response.value).trials[0]
    done = self.servicer.CompleteTrial(
        vizier_service_pb2.CompleteTrialRequest(
            name=trial.name, trial_infeasible=True,
            infeasible_reason='exploded'))
    assert done.state == study_pb2.Trial.State.Value('INFEASIBLE')
    assert done.infeasible_reason == 'exploded'

  def test_immutable_study_blocks_suggest(self):
    study = create_study(self.servicer)
    self.servicer.SetStudyState(vizier_service_pb2.SetStudyStateRequest(
        parent=study.name, state=study_pb2.Study.State.Value('COMPLETED')))
    with pytest.raises(custom_errors.ImmutableStudyError):
      self.servicer.SuggestTrials(vizier_service_pb2.SuggestTrialsRequest(
          parent=study.name, suggestion_count=1, client_id='w'))

  def test_list_optimal_trials_single_objective(self):
    study = create_study(self.servicer)
    for value in [0.1, 0.9, 0.5]:
      op = self.servicer.SuggestTrials(
          vizier_service_pb2.SuggestTrialsRequest(
              parent=study.name, suggestion_count=1, client_id='w'))
      trial = vizier_service_pb2.SuggestTrialsResponse.FromString(
          op.response.value).trials[0]
      req = vizier_service_pb2.CompleteTrialRequest(name=trial.name)
      req.final_measurement.metrics.add(metric_id='obj', value=value)
      self.servicer.CompleteTrial(req)
    resp = self.servicer.ListOptimalTrials(
        vizier_service_pb2.ListOptimalTrialsRequest(parent=study.name))
    assert len(resp.optimal_trials) == 1
    assert resp.optimal_trials[0].final_measurement.metrics[0].value == 0.9

  def test_early_stopping_flow(self):
    study = create_study(self.servicer)
    op = self.servicer.SuggestTrials(vizier_service_pb2.SuggestTrialsRequest(
        parent=study.name, suggestion_count=1, client_id='w'))
    trial = vizier_service_pb2.SuggestTrialsResponse.FromString(
        op.response.value).trials[0]
    resp = self.servicer.CheckTrialEarlyStoppingState(
        vizier_service_pb2.CheckTrialEarlyStoppingStateRequest(
            trial_name=trial.name))
    assert resp.should_stop is False  # Random policy never stops.

  def test_update_metadata_rpc(self):
    study = create_study(self.servicer)
    request = vizier_service_pb2.UpdateMetadataRequest(name=study.name)
    u = request.delta.add()
    u.metadatum.CopyFrom(study_pb2.KeyValue(key='k', ns=':a', value='v'))
    resp = self.servicer.UpdateMetadata(request)
    assert not resp.error_details
    got = self.servicer.GetStudy(vizier_service_pb2.GetStudyRequest(
        name=study.name))
    assert got.study_spec.metadata[0].value == 'v'


class TestGrpcLoopback:
  """Full client/server round trip over a real local gRPC socket."""

  @pytest.fixture(autouse=True)
  def server(self):
    self._server = DefaultVizierServer(database_url=None)
    clients.environment_variables.server_endpoint = self._server.endpoint
    yield
    clients.environment_variables.server_endpoint = 'NO_ENDPOINT'
    self._server.stop(None)

  def test_full_loop(self):
    config = make_study_config()
    study = clients.Study.from_study_config(config, owner='me',
                                            study_id='grpc_test')
    for _ in range(3):
      suggestions = study.suggest(count=2)
      assert len(suggestions) == 2
      for trial_client in suggestions:
        params = trial_client.parameters
        assert 0.0 <= params['x'] <= 1.0
        assert params['c'] in ('a', 'b')
        trial_client.complete(
            vz.Measurement(metrics={'obj': params['x']}))
    all_trials = list(study.trials().get())
    assert len(all_trials) == 6
    optimal = list(study.optimal_trials().get())
    assert len(optimal) == 1
    best_x = max(t.parameters.get_value('x') for t in all_trials)
    assert optimal[0].final_measurement.metrics['obj'].value == \
        pytest.approx(best_x)

  def test_study_config_roundtrip_through_server(self):
    config = make_study_config()
    study = clients.Study.from_study_config(config, owner='me',
                                            study_id='roundtrip')
    materialized = study.materialize_study_config()
    assert materialized.search_space == config.search_space
    assert materialized.algorithm == config.algorithm

  def test_state_transitions(self):
    study = clients.Study.from_study_config(make_study_config(), owner='me',
                                            study_id='state_test')
    assert study.materialize_state() == vz.StudyState.ACTIVE
    study.set_state(vz.StudyState.COMPLETED)
    assert study.materialize_state() == vz.StudyState.COMPLETED
    # Suggest on a completed study returns no trials.
    assert study.suggest(count=1) == []


class TestDistributedPythia:

  def test_suggest_through_remote_pythia(self):
    server = DistributedPythiaVizierServer(database_url=None)
    try:
      clients.environment_variables.server_endpoint = server.endpoint
      study = clients.Study.from_study_config(
          make_study_config(), owner='me', study_id='dist')
      suggestions = study.suggest(count=2)
      assert len(suggestions) == 2
    finally:
      clients.environment_variables.server_endpoint = 'NO_ENDPOINT'
      server.stop(None)


class TestMultiClientStress:
  """Thread-pool stress: many clients against one in-process server."""

  def test_concurrent_clients_consistent_ids(self):
    servicer = VizierServicer(database_url=None)
    study = create_study(servicer)
    num_clients, per_client = 8, 4
    errors = []

    def run_client(idx):
      try:
        for _ in range(per_client):
          op = servicer.SuggestTrials(
              vizier_service_pb2.SuggestTrialsRequest(
                  parent=study.name, suggestion_count=1,
                  client_id=f'client{idx}'))
          trial = vizier_service_pb2.SuggestTrialsResponse.FromString(
              op.response.value).trials[0]
          req = vizier_service_pb2.CompleteTrialRequest(name=trial.name)
          req.final_measurement.metrics.add(metric_id='obj', value=0.5)
          servicer.CompleteTrial(req)
      except Exception as e:  # pragma: no cover
        errors.append(e)

    threads = [threading.Thread(target=run_client, args=(i,))
               for i in range(num_clients)]
    for t in threads:
      t.start()
    for t in threads:
      t.join()
    assert not errors
    trials = servicer.ListTrials(vizier_service_pb2.ListTrialsRequest(
        parent=study.name)).trials
    assert len(trials) == num_clients * per_client
    ids = sorted(int(t.id) for t in trials)
    assert ids == list(range(1, num_clients * per_client + 1))

  def test_hundred_concurrent_clients_thread_pool(self):
    """Reference performance_test.py:44-92 scale: 100 clients via a
    ThreadPool against one server, trial-id consistency + latency log."""
    import multiprocessing.pool
    import time as _time

    servicer = VizierServicer(database_url=None)
    study = create_study(servicer, display_name='stress100')
    num_clients = 100

    def run_client(idx):
      t0 = _time.monotonic()
      op = servicer.SuggestTrials(
          vizier_service_pb2.SuggestTrialsRequest(
              parent=study.name, suggestion_count=1,
              client_id=f'c{idx}'))
      trial = vizier_service_pb2.SuggestTrialsResponse.FromString(
          op.response.value).trials[0]
      req = vizier_service_pb2.CompleteTrialRequest(name=trial.name)
      req.final_measurement.metrics.add(metric_id='obj', value=0.1)
      servicer.CompleteTrial(req)
      return _time.monotonic() - t0

    with multiprocessing.pool.ThreadPool(16) as pool:
      latencies = pool.map(run_client, range(num_clients))

    trials = servicer.ListTrials(vizier_service_pb2.ListTrialsRequest(
        parent=study.name)).trials
    assert len(trials) == num_clients
    ids = sorted(int(t.id) for t in trials)
    assert ids == list(range(1, num_clients + 1))
    # Latency sanity (the only latency logging, like the reference).
    import numpy as _np
    print(f'suggest+complete latency: p50={_np.median(latencies)*1e3:.1f}'
          f'ms p95={_np.percentile(latencies, 95)*1e3:.1f}ms')


class TestClientAbcConformance(
    __import__('vizier_amd.client.client_abc_testing',
               fromlist=['TestCaseMixin']).TestCaseMixin):
  """Runs the client ABC conformance mixin against clients.Study
  (reference client_abc_testing.py:36-48 pattern)."""

  def create_study(self, problem, study_id: str):
    from vizier_amd._src.service import clients as service_clients
    from vizier_amd._src.service import constants
    config = vz.StudyConfig(algorithm='RANDOM_SEARCH')
    for pc in problem.search_space.parameters:
      config.search_space.add(pc)
    for mi in problem.metric_information:
      config.metric_information.append(mi)
    service_clients.environment_variables.server_endpoint = \
        constants.NO_ENDPOINT
    return service_clients.Study.from_study_config(
        config, owner='conformance', study_id=study_id)
