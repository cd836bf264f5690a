"""Differential fuzzing: RAM vs SQL datastores must agree exactly.

Random operation sequences are applied to both implementations; after
every operation the observable state (studies, trials, operations,
errors raised) must be identical. This is the strongest form of the
contract test: any divergence between the in-memory tree and the
SQLAlchemy tables shows up as a concrete op sequence.
"""

import random

import pytest

from vizier_amd._src.service import custom_errors
from vizier_amd._src.service.proto import (study_pb2,
                                           vizier_oss_pb2,
                                           vizier_service_pb2)
from vizier_amd._src.service.ram_datastore import NestedDictRAMDataStore


def make_sql_store():
  import sqlalchemy as sqla
  from vizier_amd._src.service.sql_datastore import SQLDataStore
  return SQLDataStore(sqla.create_engine('sqlite:///:memory:'))


def make_study(owner, name):
  study = study_pb2.Study(
      name=f'owners/{owner}/studies/{name}', display_name=name)
  spec = study.study_spec
  p = spec.parameters.add()
  p.parameter_id = 'x'
  p.double_value_spec.min_value = 0.0
  p.double_value_spec.max_value = 1.0
  m = spec.metrics.add()
  m.metric_id = 'obj'
  return study


def make_trial(study_name, tid, value=0.5):
  t = study_pb2.Trial(name=f'{study_name}/trials/{tid}', id=str(tid))
  t.state = study_pb2.Trial.State.Value('ACTIVE')
  pr = t.parameters.add()
  pr.parameter_id = 'x'
  pr.value.number_value = value
  return t


def make_sugg_op(study_name, client, number):
  from vizier_amd._src.service.proto import operations_pb2
  owner = study_name.split('/')[1]
  study = study_name.split('/')[3]
  return operations_pb2.Operation(
      name=f'owners/{owner}/operations/suggestion/{study}/{client}/'
           f'{number}')


def make_es_op(study_name, number):
  owner = study_name.split('/')[1]
  study = study_name.split('/')[3]
  return vizier_oss_pb2.EarlyStoppingOperation(
      name=f'owners/{owner}/operations/earlystopping/{study}/{number}')


def apply(store, op):
  """Applies one op; returns a comparable (result, error) pair."""
  kind = op[0]
  try:
    if kind == 'create_study':
      return store.create_study(make_study(op[1], op[2])), None
    if kind == 'load_study':
      return store.load_study(op[1]).SerializeToString(), None
    if kind == 'delete_study':
      return store.delete_study(op[1]), None
    if kind == 'list_studies':
      return sorted(s.name for s in store.list_studies(op[1])), None
    if kind == 'create_trial':
      return store.create_trial(make_trial(op[1], op[2], op[3])), None
    if kind == 'get_trial':
      return store.get_trial(op[1]).SerializeToString(), None
    if kind == 'list_trials':
      return sorted(t.name for t in store.list_trials(op[1])), None
    if kind == 'delete_trial':
      return store.delete_trial(op[1]), None
    if kind == 'max_trial_id':
      return store.max_trial_id(op[1]), None
    if kind == 'create_sugg_op':
      return store.create_suggestion_operation(
          make_sugg_op(op[1], op[2], op[3])), None
    if kind == 'get_sugg_op':
      return store.get_suggestion_operation(op[1]).SerializeToString(), \
          None
    if kind == 'max_sugg_number':
      return store.max_suggestion_operation_number(op[1], op[2]), None
    if kind == 'create_es_op':
      return store.create_early_stopping_operation(
          make_es_op(op[1], op[2])), None
    if kind == 'get_es_op':
      return store.get_early_stopping_operation(
          op[1]).SerializeToString(), None
    if kind == 'list_sugg_ops':
      return sorted(
          o.name for o in store.list_suggestion_operations(
              op[1], op[2])), None
    if kind == 'update_metadata':
      kv = study_pb2.KeyValue(key=op[2], ns=':fuzz', value=op[3])
      store.update_metadata(op[1], [kv], [])
      return store.load_study(op[1]).SerializeToString(), None
    if kind == 'update_trial_metadata':
      u = vizier_service_pb2.UnitMetadataUpdate(trial_id=str(op[2]))
      u.metadatum.key = op[3]
      u.metadatum.ns = ':fuzz'
      u.metadatum.value = op[4]
      store.update_metadata(op[1], [], [u])
      return store.get_trial(
          f'{op[1]}/trials/{op[2]}').SerializeToString(), None
    raise AssertionError(op)
  except custom_errors.NotFoundError:
    return None, 'NotFoundError'
  except custom_errors.AlreadyExistsError:
    return None, 'AlreadyExistsError'
  except custom_errors.ImmutableStudyError:
    return None, 'ImmutableStudyError'
  except KeyError:
    # Trial-metadata update on a missing trial raises KeyError in the
    # reference's ram_datastore; both stores must match it.
    return None, 'KeyError'


def random_op(rng, owners, studies, trial_ids):
  choice = rng.choice([
      'create_study', 'load_study', 'list_studies', 'create_trial',
      'get_trial', 'list_trials', 'max_trial_id', 'delete_trial',
      'delete_study', 'create_sugg_op', 'get_sugg_op',
      'max_sugg_number', 'create_es_op', 'get_es_op',
      'list_sugg_ops', 'update_metadata', 'update_trial_metadata'])
  owner = rng.choice(owners)
  study = rng.choice(studies)
  study_name = f'owners/{owner}/studies/{study}'
  if choice == 'create_study':
    return ('create_study', owner, study)
  if choice == 'load_study':
    return ('load_study', study_name)
  if choice == 'delete_study':
    return ('delete_study', study_name)
  if choice == 'list_studies':
    return ('list_studies', f'owners/{owner}')
  tid = rng.choice(trial_ids)
  if choice == 'create_trial':
    return ('create_trial', study_name, tid, rng.random())
  if choice == 'get_trial':
    return ('get_trial', f'{study_name}/trials/{tid}')
  if choice == 'delete_trial':
    return ('delete_trial', f'{study_name}/trials/{tid}')
  if choice == 'list_trials':
    return ('list_trials', study_name)
  if choice == 'create_sugg_op':
    return ('create_sugg_op', study_name, f'cl{rng.randint(0, 1)}',
            rng.randint(1, 3))
  if choice == 'get_sugg_op':
    owner, study = study_name.split('/')[1], study_name.split('/')[3]
    return ('get_sugg_op',
            f'owners/{owner}/operations/suggestion/{study}/'
            f'cl{rng.randint(0, 1)}/{rng.randint(1, 3)}')
  if choice == 'max_sugg_number':
    return ('max_sugg_number', study_name, f'cl{rng.randint(0, 1)}')
  if choice == 'create_es_op':
    return ('create_es_op', study_name, rng.randint(1, 3))
  if choice == 'get_es_op':
    owner, study = study_name.split('/')[1], study_name.split('/')[3]
    return ('get_es_op',
            f'owners/{owner}/operations/earlystopping/{study}/'
            f'{rng.randint(1, 3)}')
  if choice == 'list_sugg_ops':
    return ('list_sugg_ops', study_name, f'cl{rng.randint(0, 1)}')
  if choice == 'update_metadata':
    return ('update_metadata', study_name, f'k{rng.randint(0, 2)}',
            f'v{rng.randint(0, 9)}')
  if choice == 'update_trial_metadata':
    return ('update_trial_metadata', study_name, tid,
            f'k{rng.randint(0, 2)}', f'v{rng.randint(0, 9)}')
  return ('max_trial_id', study_name)


@pytest.mark.parametrize('seed', range(6))
def test_ram_and_sql_stores_agree_on_random_sequences(seed):
  rng = random.Random(seed)
  ram = NestedDictRAMDataStore()
  sql = make_sql_store()
  owners = ['o1', 'o2']
  studies = ['s1', 's2', 's3']
  trial_ids = [1, 2, 3, 4]
  history = []
  for step in range(120):
    op = random_op(rng, owners, studies, trial_ids)
    history.append(op)
    got_ram = apply(ram, op)
    got_sql = apply(sql, op)
    assert got_ram == got_sql, (
        f'step {step}: {op} diverged:\n ram={got_ram}\n sql={got_sql}\n'
        f'history={history[-8:]}')
