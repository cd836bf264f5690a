"""Contract tests run against both datastore implementations."""

import pytest
import sqlalchemy as sqla

from vizier_amd._src.service import custom_errors, resources
from vizier_amd._src.service.ram_datastore import NestedDictRAMDataStore
from vizier_amd._src.service.sql_datastore import SQLDataStore
from vizier_amd._src.service.proto import (
    operations_pb2,
    study_pb2,
    vizier_oss_pb2,
    vizier_service_pb2,
)

OWNER = 'owners/alice'
STUDY = 'owners/alice/studies/s1'


def make_study(name=STUDY):
  study = study_pb2.Study(name=name, display_name=name.split('/')[-1])
  study.study_spec.algorithm = 'RANDOM_SEARCH'
  return study


def make_trial(study_name=STUDY, trial_id=1):
  r = resources.StudyResource.from_name(study_name)
  t = study_pb2.Trial(id=str(trial_id),
                      name=r.trial_resource(trial_id).name)
  t.state = study_pb2.Trial.State.Value('ACTIVE')
  return t


@pytest.fixture(params=['ram', 'sql'])
def ds(request):
  if request.param == 'ram':
    return NestedDictRAMDataStore()
  engine = sqla.create_engine('sqlite:///:memory:',
                              connect_args={'check_same_thread': False},
                              poolclass=sqla.pool.StaticPool)
  return SQLDataStore(engine)


class TestDataStoreContract:

  def test_study_crud(self, ds):
    study = make_study()
    assert ds.create_study(study) == STUDY
    assert ds.load_study(STUDY) == study
    with pytest.raises(custom_errors.AlreadyExistsError):
      ds.create_study(study)
    study2 = make_study()
    study2.state = study_pb2.Study.State.Value('COMPLETED')
    ds.update_study(study2)
    assert ds.load_study(STUDY).state == \
        study_pb2.Study.State.Value('COMPLETED')
    assert len(ds.list_studies(OWNER)) == 1
    ds.delete_study(STUDY)
    with pytest.raises(custom_errors.NotFoundError):
      ds.load_study(STUDY)
    with pytest.raises(custom_errors.NotFoundError):
      ds.list_studies(OWNER)

  def test_pass_by_value(self, ds):
    study = make_study()
    ds.create_study(study)
    study.display_name = 'mutated-after-store'
    assert ds.load_study(STUDY).display_name == 's1'

  def test_trial_crud(self, ds):
    ds.create_study(make_study())
    t1, t2 = make_trial(trial_id=1), make_trial(trial_id=2)
    ds.create_trial(t1)
    ds.create_trial(t2)
    assert ds.max_trial_id(STUDY) == 2
    assert [t.id for t in ds.list_trials(STUDY)] == ['1', '2']
    got = ds.get_trial(t1.name)
    assert got == t1
    t1.state = study_pb2.Trial.State.Value('SUCCEEDED')
    ds.update_trial(t1)
    assert ds.get_trial(t1.name).state == \
        study_pb2.Trial.State.Value('SUCCEEDED')
    ds.delete_trial(t2.name)
    assert len(ds.list_trials(STUDY)) == 1
    with pytest.raises(custom_errors.NotFoundError):
      ds.get_trial(t2.name)

  def test_suggestion_operations(self, ds):
    ds.create_study(make_study())
    r = resources.SuggestionOperationResource('alice', 's1', 'c1', 1)
    op = operations_pb2.Operation(name=r.name, done=False)
    ds.create_suggestion_operation(op)
    with pytest.raises(custom_errors.NotFoundError):
      ds.max_suggestion_operation_number(STUDY, 'unknown_client')
    assert ds.max_suggestion_operation_number(STUDY, 'c1') == 1
    got = ds.get_suggestion_operation(r.name)
    assert got == op
    op.done = True
    ds.update_suggestion_operation(op)
    ops = ds.list_suggestion_operations(STUDY, 'c1')
    assert len(ops) == 1 and ops[0].done
    ops = ds.list_suggestion_operations(STUDY, 'c1', lambda o: not o.done)
    assert not ops

  def test_early_stopping_operations(self, ds):
    ds.create_study(make_study())
    ds.create_trial(make_trial(trial_id=1))
    r = resources.EarlyStoppingOperationResource('alice', 's1', 1)
    op = vizier_oss_pb2.EarlyStoppingOperation(
        name=r.name,
        status=vizier_oss_pb2.EarlyStoppingOperation.Status.Value('ACTIVE'))
    ds.create_early_stopping_operation(op)
    assert ds.get_early_stopping_operation(r.name) == op
    op.should_stop = True
    ds.update_early_stopping_operation(op)
    assert ds.get_early_stopping_operation(r.name).should_stop

  def test_update_metadata(self, ds):
    ds.create_study(make_study())
    ds.create_trial(make_trial(trial_id=1))
    kv = study_pb2.KeyValue(key='k', ns=':algo', value='v')
    update = vizier_service_pb2.UnitMetadataUpdate(trial_id='1')
    update.metadatum.CopyFrom(study_pb2.KeyValue(key='tk', value='tv'))
    ds.update_metadata(STUDY, [kv], [update])
    study = ds.load_study(STUDY)
    assert study.study_spec.metadata[0].key == 'k'
    trial = ds.get_trial(make_trial(trial_id=1).name)
    assert trial.metadata[0].key == 'tk'
    # Unknown trial -> KeyError.
    bad = vizier_service_pb2.UnitMetadataUpdate(trial_id='99')
    bad.metadatum.CopyFrom(study_pb2.KeyValue(key='x', value='y'))
    with pytest.raises(KeyError):
      ds.update_metadata(STUDY, [], [bad])
