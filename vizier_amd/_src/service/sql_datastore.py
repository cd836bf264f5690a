"""SQLAlchemy-backed datastore storing serialized protos.

Capability parity with vizier/_src/service/sql_datastore.py:40 (five
tables keyed by resource name, values are serialized protos). Works with
SQLite (file or :memory:) and any SQLAlchemy 2.0 engine.
"""

from __future__ import annotations

from typing import Callable, Iterable, List, Optional

import sqlalchemy as sqla

from vizier_amd._src.pyvizier import metadata_util
from vizier_amd._src.service import custom_errors, resources
from vizier_amd._src.service.datastore import DataStore
from vizier_amd._src.service.proto import (
    operations_pb2,
    study_pb2,
    vizier_oss_pb2,
)


class SQLDataStore(DataStore):
  """Studies/trials/operations in SQL tables of serialized protos."""

  def __init__(self, engine: sqla.engine.Engine):
    self._engine = engine
    self._md = sqla.MetaData()
    self._studies = sqla.Table(
        'studies', self._md,
        sqla.Column('study_name', sqla.String, primary_key=True),
        sqla.Column('owner_name', sqla.String, index=True),
        sqla.Column('serialized_study', sqla.LargeBinary))
    self._trials = sqla.Table(
        'trials', self._md,
        sqla.Column('trial_name', sqla.String, primary_key=True),
        sqla.Column('study_name', sqla.String, index=True),
        sqla.Column('trial_id', sqla.Integer),
        sqla.Column('serialized_trial', sqla.LargeBinary))
    self._suggestion_ops = sqla.Table(
        'suggestion_operations', self._md,
        sqla.Column('operation_name', sqla.String, primary_key=True),
        sqla.Column('study_name', sqla.String, index=True),
        sqla.Column('client_id', sqla.String),
        sqla.Column('operation_number', sqla.Integer),
        sqla.Column('serialized_op', sqla.LargeBinary))
    self._early_stopping_ops = sqla.Table(
        'early_stopping_operations', self._md,
        sqla.Column('operation_name', sqla.String, primary_key=True),
        sqla.Column('study_name', sqla.String, index=True),
        sqla.Column('serialized_op', sqla.LargeBinary))
    self._md.create_all(self._engine)

  # -- studies --------------------------------------------------------------

  def create_study(self, study) -> str:
    r = resources.StudyResource.from_name(study.name)
    with self._engine.begin() as conn:
      exists = conn.execute(sqla.select(self._studies.c.study_name).where(
          self._studies.c.study_name == study.name)).first()
      if exists:
        raise custom_errors.AlreadyExistsError(
            f'Study already exists: {study.name}')
      conn.execute(self._studies.insert().values(
          study_name=study.name, owner_name=r.owner_resource.name,
          serialized_study=study.SerializeToString()))
    return study.name

  def load_study(self, study_name: str):
    with self._engine.begin() as conn:
      row = conn.execute(sqla.select(self._studies.c.serialized_study).where(
          self._studies.c.study_name == study_name)).first()
    if row is None:
      raise custom_errors.NotFoundError(f'Study not found: {study_name}')
    return study_pb2.Study.FromString(row[0])

  def update_study(self, study) -> str:
    with self._engine.begin() as conn:
      result = conn.execute(self._studies.update().where(
          self._studies.c.study_name == study.name).values(
              serialized_study=study.SerializeToString()))
      if result.rowcount == 0:
        raise custom_errors.NotFoundError(f'Study not found: {study.name}')
    return study.name

  def delete_study(self, study_name: str) -> None:
    with self._engine.begin() as conn:
      result = conn.execute(self._studies.delete().where(
          self._studies.c.study_name == study_name))
      if result.rowcount == 0:
        raise custom_errors.NotFoundError(f'Study not found: {study_name}')
      conn.execute(self._trials.delete().where(
          self._trials.c.study_name == study_name))
      conn.execute(self._suggestion_ops.delete().where(
          self._suggestion_ops.c.study_name == study_name))
      conn.execute(self._early_stopping_ops.delete().where(
          self._early_stopping_ops.c.study_name == study_name))

  def list_studies(self, owner_name: str) -> List:
    with self._engine.begin() as conn:
      rows = conn.execute(sqla.select(self._studies.c.serialized_study).where(
          self._studies.c.owner_name == owner_name)).all()
    if not rows:
      raise custom_errors.NotFoundError(f'Owner not found: {owner_name}')
    return [study_pb2.Study.FromString(row[0]) for row in rows]

  # -- trials ---------------------------------------------------------------

  def _study_exists(self, conn, study_name: str) -> bool:
    return conn.execute(sqla.select(self._studies.c.study_name).where(
        self._studies.c.study_name == study_name)).first() is not None

  def create_trial(self, trial) -> str:
    r = resources.TrialResource.from_name(trial.name)
    with self._engine.begin() as conn:
      if not self._study_exists(conn, r.study_resource.name):
        raise custom_errors.NotFoundError(
            f'Study not found: {r.study_resource.name}')
      exists = conn.execute(sqla.select(self._trials.c.trial_name).where(
          self._trials.c.trial_name == trial.name)).first()
      if exists:
        raise custom_errors.AlreadyExistsError(
            f'Trial already exists: {trial.name}')
      conn.execute(self._trials.insert().values(
          trial_name=trial.name, study_name=r.study_resource.name,
          trial_id=r.trial_id, serialized_trial=trial.SerializeToString()))
    return trial.name

  def get_trial(self, trial_name: str):
    with self._engine.begin() as conn:
      row = conn.execute(sqla.select(self._trials.c.serialized_trial).where(
          self._trials.c.trial_name == trial_name)).first()
    if row is None:
      raise custom_errors.NotFoundError(f'Trial not found: {trial_name}')
    return study_pb2.Trial.FromString(row[0])

  def update_trial(self, trial) -> str:
    with self._engine.begin() as conn:
      result = conn.execute(self._trials.update().where(
          self._trials.c.trial_name == trial.name).values(
              serialized_trial=trial.SerializeToString()))
      if result.rowcount == 0:
        raise custom_errors.NotFoundError(f'Trial not found: {trial.name}')
    return trial.name

  def list_trials(self, study_name: str) -> List:
    with self._engine.begin() as conn:
      if not self._study_exists(conn, study_name):
        raise custom_errors.NotFoundError(f'Study not found: {study_name}')
      rows = conn.execute(
          sqla.select(self._trials.c.serialized_trial).where(
              self._trials.c.study_name == study_name).order_by(
                  self._trials.c.trial_id)).all()
    return [study_pb2.Trial.FromString(row[0]) for row in rows]

  def delete_trial(self, trial_name: str) -> None:
    with self._engine.begin() as conn:
      result = conn.execute(self._trials.delete().where(
          self._trials.c.trial_name == trial_name))
      if result.rowcount == 0:
        raise custom_errors.NotFoundError(f'Trial not found: {trial_name}')

  def max_trial_id(self, study_name: str) -> int:
    with self._engine.begin() as conn:
      exists = conn.execute(sqla.select(self._studies.c.study_name).where(
          self._studies.c.study_name == study_name)).first()
      if exists is None:
        # Reference parity (sql_datastore.py:335): missing study raises.
        raise custom_errors.NotFoundError(
            f'Study {study_name} does not exist.')
      row = conn.execute(sqla.select(
          sqla.func.max(self._trials.c.trial_id)).where(
              self._trials.c.study_name == study_name)).first()
    return int(row[0]) if row and row[0] is not None else 0

  # -- suggestion operations ------------------------------------------------

  def create_suggestion_operation(self, operation) -> str:
    r = resources.SuggestionOperationResource.from_name(operation.name)
    study_name = resources.StudyResource(r.owner_id, r.study_id).name
    with self._engine.begin() as conn:
      if not self._study_exists(conn, study_name):
        raise custom_errors.NotFoundError(f'Study not found: {study_name}')
      exists = conn.execute(
          sqla.select(self._suggestion_ops.c.operation_name).where(
              self._suggestion_ops.c.operation_name == operation.name)
      ).first()
      if exists:
        raise custom_errors.AlreadyExistsError(
            f'Operation already exists: {operation.name}')
      conn.execute(self._suggestion_ops.insert().values(
          operation_name=operation.name, study_name=study_name,
          client_id=r.client_id, operation_number=r.operation_number,
          serialized_op=operation.SerializeToString()))
    return operation.name

  def get_suggestion_operation(self, operation_name: str):
    with self._engine.begin() as conn:
      row = conn.execute(
          sqla.select(self._suggestion_ops.c.serialized_op).where(
              self._suggestion_ops.c.operation_name == operation_name)
      ).first()
    if row is None:
      raise custom_errors.NotFoundError(
          f'Operation not found: {operation_name}')
    return operations_pb2.Operation.FromString(row[0])

  def update_suggestion_operation(self, operation) -> str:
    with self._engine.begin() as conn:
      result = conn.execute(self._suggestion_ops.update().where(
          self._suggestion_ops.c.operation_name == operation.name).values(
              serialized_op=operation.SerializeToString()))
      if result.rowcount == 0:
        raise custom_errors.NotFoundError(
            f'Operation not found: {operation.name}')
    return operation.name

  def list_suggestion_operations(
      self, study_name: str, client_id: str,
      filter_fn: Optional[Callable[[object], bool]] = None) -> List:
    with self._engine.begin() as conn:
      rows = conn.execute(
          sqla.select(self._suggestion_ops.c.serialized_op).where(
              sqla.and_(self._suggestion_ops.c.study_name == study_name,
                        self._suggestion_ops.c.client_id == client_id)
          ).order_by(self._suggestion_ops.c.operation_number)).all()
    if not rows:
      raise custom_errors.NotFoundError(
          f'No operations for client: {client_id}')
    ops = [operations_pb2.Operation.FromString(row[0]) for row in rows]
    if filter_fn is not None:
      ops = [op for op in ops if filter_fn(op)]
    return ops

  def max_suggestion_operation_number(self, study_name: str,
                                      client_id: str) -> int:
    with self._engine.begin() as conn:
      row = conn.execute(sqla.select(
          sqla.func.max(self._suggestion_ops.c.operation_number)).where(
              sqla.and_(self._suggestion_ops.c.study_name == study_name,
                        self._suggestion_ops.c.client_id == client_id))
      ).first()
    if row is None or row[0] is None:
      raise custom_errors.NotFoundError(
          f'No operations for client: {client_id}')
    return int(row[0])

  # -- early-stopping operations -------------------------------------------

  def create_early_stopping_operation(self, operation) -> str:
    r = resources.EarlyStoppingOperationResource.from_name(operation.name)
    study_name = resources.StudyResource(r.owner_id, r.study_id).name
    with self._engine.begin() as conn:
      if not self._study_exists(conn, study_name):
        raise custom_errors.NotFoundError(f'Study not found: {study_name}')
      exists = conn.execute(
          sqla.select(self._early_stopping_ops.c.operation_name).where(
              self._early_stopping_ops.c.operation_name == operation.name)
      ).first()
      if exists:
        raise custom_errors.AlreadyExistsError(
            f'Operation already exists: {operation.name}')
      conn.execute(self._early_stopping_ops.insert().values(
          operation_name=operation.name, study_name=study_name,
          serialized_op=operation.SerializeToString()))
    return operation.name

  def get_early_stopping_operation(self, operation_name: str):
    with self._engine.begin() as conn:
      row = conn.execute(
          sqla.select(self._early_stopping_ops.c.serialized_op).where(
              self._early_stopping_ops.c.operation_name == operation_name)
      ).first()
    if row is None:
      raise custom_errors.NotFoundError(
          f'Operation not found: {operation_name}')
    return vizier_oss_pb2.EarlyStoppingOperation.FromString(row[0])

  def update_early_stopping_operation(self, operation) -> str:
    with self._engine.begin() as conn:
      result = conn.execute(self._early_stopping_ops.update().where(
          self._early_stopping_ops.c.operation_name == operation.name
      ).values(serialized_op=operation.SerializeToString()))
      if result.rowcount == 0:
        raise custom_errors.NotFoundError(
            f'Operation not found: {operation.name}')
    return operation.name

  # -- metadata -------------------------------------------------------------

  def update_metadata(self, study_name: str, study_metadata: Iterable,
                      trial_metadata: Iterable) -> None:
    study = self.load_study(study_name)  # NotFoundError if missing
    metadata_util.merge_study_metadata(study.study_spec, study_metadata)
    self.update_study(study)
    r = resources.StudyResource.from_name(study_name)
    for update in trial_metadata:
      trial_name = r.trial_resource(int(update.trial_id)).name
      try:
        trial = self.get_trial(trial_name)
      except custom_errors.NotFoundError:
        raise KeyError(f'Trial {update.trial_id} not found in {study_name}')
      metadata_util.merge_trial_metadata(trial, [update.metadatum])
      self.update_trial(trial)
