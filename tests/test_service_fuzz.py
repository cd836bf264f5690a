"""Randomized servicer workloads: invariants under arbitrary interleavings.

Complements the scripted state-machine tests in test_service_e2e.py:
random sequences of suggest/complete/stop/add-measurement/delete from
several simulated clients must never violate the service's structural
invariants (unique dense trial ids, per-client ACTIVE reuse, final
measurements on SUCCEEDED trials, REQUESTED-pool draining). Mirrors the
reference's thread-pool stress approach (performance_test.py) but with
randomized operation ORDER rather than concurrency.
"""

import random

import pytest

from vizier_amd._src.service import custom_errors
from vizier_amd._src.service.proto import study_pb2, vizier_service_pb2
from vizier_amd._src.service.vizier_service import VizierServicer
from tests.test_service_e2e import create_study

_STATE = study_pb2.Trial.State


def _suggest(servicer, study_name, client_id, count=1):
  op = servicer.SuggestTrials(vizier_service_pb2.SuggestTrialsRequest(
      parent=study_name, suggestion_count=count, client_id=client_id))
  assert op.done, op
  if op.HasField('error'):
    return []
  resp = vizier_service_pb2.SuggestTrialsResponse.FromString(
      op.response.value)
  return list(resp.trials)


def _list_trials(servicer, study_name):
  return list(servicer.ListTrials(
      vizier_service_pb2.ListTrialsRequest(parent=study_name)).trials)


@pytest.mark.parametrize('seed', range(4))
def test_randomized_single_study_workload(seed):
  rng = random.Random(seed)
  servicer = VizierServicer(database_url=None)
  study = create_study(servicer, display_name=f'fuzz{seed}')
  clients = [f'worker{i}' for i in range(3)]
  # Track what each client believes it owns (ACTIVE trial names).
  completed = set()

  for _ in range(60):
    action = rng.choice(['suggest', 'complete', 'add_measurement',
                         'stop', 'create_requested'])
    client = rng.choice(clients)
    if action == 'suggest':
      count = rng.randint(1, 3)
      trials = _suggest(servicer, study.name, client, count)
      assert len(trials) >= 1
      for t in trials:
        assert t.state == _STATE.Value('ACTIVE')
        assert t.client_id == client
      # Immediately re-suggesting for the SAME client must return the
      # SAME active trials (idempotent assignment), not new ones.
      again = _suggest(servicer, study.name, client, count)
      assert sorted(t.name for t in again) == \
          sorted(t.name for t in trials)
    elif action == 'create_requested':
      # A REQUESTED trial enters the pool and must be handed out by a
      # later suggest before any fresh Pythia trials.
      t = study_pb2.Trial(state=_STATE.Value('REQUESTED'))
      p = t.parameters.add()
      p.parameter_id = 'x'
      p.value.number_value = rng.random()
      servicer.CreateTrial(vizier_service_pb2.CreateTrialRequest(
          parent=study.name, trial=t))
    else:
      active = [t for t in _list_trials(servicer, study.name)
                if t.state == _STATE.Value('ACTIVE')
                and t.client_id == client]
      if not active:
        continue
      target = rng.choice(active)
      if action == 'add_measurement':
        req = vizier_service_pb2.AddTrialMeasurementRequest(
            trial_name=target.name)
        req.measurement.metrics.add(metric_id='obj',
                                    value=rng.random())
        servicer.AddTrialMeasurement(req)
      elif action == 'complete':
        req = vizier_service_pb2.CompleteTrialRequest(name=target.name)
        roll = rng.random()
        if roll < 0.6:
          req.final_measurement.metrics.add(metric_id='obj',
                                            value=rng.random())
        elif roll < 0.8:
          req.trial_infeasible = True
          req.infeasible_reason = 'fuzz-infeasible'
        # else: no final measurement — the service must select the last
        # intermediate one, or REJECT if there are none (reference
        # behavior: ValueError when neither is present).
        try:
          got = servicer.CompleteTrial(req)
        except custom_errors.ImmutableTrialError:
          continue
        except ValueError:
          # Only legal for the fully-empty case: no final measurement
          # in the request, no infeasible_reason, and no intermediate
          # measurements on the trial.
          assert not req.HasField('final_measurement')
          assert not req.trial_infeasible
          assert not list(target.measurements)
          continue
        completed.add(got.name)
        assert got.state in (_STATE.Value('SUCCEEDED'),
                             _STATE.Value('INFEASIBLE'))
        if got.state == _STATE.Value('SUCCEEDED'):
          assert got.final_measurement.metrics
      elif action == 'stop':
        servicer.StopTrial(vizier_service_pb2.StopTrialRequest(
            name=target.name))

    # --- Invariants, checked after every operation. ---
    trials = _list_trials(servicer, study.name)
    names = [t.name for t in trials]
    assert len(names) == len(set(names)), 'duplicate trial names'
    ids = sorted(int(t.id) for t in trials)
    assert ids == list(range(1, len(ids) + 1)), f'ids not dense: {ids}'
    for t in trials:
      if t.name in completed and t.state == _STATE.Value('SUCCEEDED'):
        assert t.final_measurement.metrics
      if t.state == _STATE.Value('ACTIVE'):
        assert t.client_id, 'ACTIVE trial with no owner'


@pytest.mark.parametrize('seed', range(2))
def test_randomized_multi_study_isolation(seed):
  """Operations on one study never leak into another."""
  rng = random.Random(100 + seed)
  servicer = VizierServicer(database_url=None)
  studies = [create_study(servicer, owner=f'o{i % 2}',
                          display_name=f's{i}').name for i in range(3)]
  counts = {s: 0 for s in studies}
  for _ in range(40):
    s = rng.choice(studies)
    got = _suggest(servicer, s, f'cl{rng.randint(0, 1)}',
                   rng.randint(1, 2))
    for t in got:
      assert t.name.startswith(s + '/trials/')
    # Complete everything so the next suggest makes fresh trials.
    for t in got:
      req = vizier_service_pb2.CompleteTrialRequest(name=t.name)
      req.final_measurement.metrics.add(metric_id='obj', value=0.5)
      try:
        servicer.CompleteTrial(req)
      except custom_errors.ImmutableTrialError:
        pass
    counts[s] = len(_list_trials(servicer, s))
  for s in studies:
    assert len(_list_trials(servicer, s)) == counts[s]
