"""In-RAM datastore.

Capability parity with vizier/_src/service/ram_datastore.py:83
(NestedDictRAMDataStore), with a single RLock guarding the nested dicts.
"""

from __future__ import annotations

import copy
import threading
from typing import Callable, Dict, Iterable, List, Optional

from vizier_amd._src.pyvizier import metadata_util
from vizier_amd._src.service import custom_errors, resources
from vizier_amd._src.service.datastore import DataStore


class _StudyNode:

  def __init__(self, study_proto):
    self.study_proto = study_proto
    self.trials: Dict[int, object] = {}
    # client_id -> {operation_number: Operation}
    self.suggestion_ops: Dict[str, Dict[int, object]] = {}
    # operation_id -> EarlyStoppingOperation
    self.early_stopping_ops: Dict[str, object] = {}


class NestedDictRAMDataStore(DataStore):
  """owner -> study -> {trials, suggestion ops, early-stopping ops}."""

  def __init__(self):
    self._owners: Dict[str, Dict[str, _StudyNode]] = {}
    self._lock = threading.RLock()

  # -- helpers --------------------------------------------------------------

  def _study_node(self, study_name: str) -> _StudyNode:
    r = resources.StudyResource.from_name(study_name)
    try:
      return self._owners[r.owner_id][r.study_id]
    except KeyError:
      raise custom_errors.NotFoundError(f'Study not found: {study_name}')

  # -- studies --------------------------------------------------------------

  def create_study(self, study) -> str:
    r = resources.StudyResource.from_name(study.name)
    with self._lock:
      studies = self._owners.setdefault(r.owner_id, {})
      if r.study_id in studies:
        raise custom_errors.AlreadyExistsError(
            f'Study already exists: {study.name}')
      studies[r.study_id] = _StudyNode(copy.deepcopy(study))
    return study.name

  def load_study(self, study_name: str):
    with self._lock:
      return copy.deepcopy(self._study_node(study_name).study_proto)

  def update_study(self, study) -> str:
    with self._lock:
      node = self._study_node(study.name)
      node.study_proto = copy.deepcopy(study)
    return study.name

  def delete_study(self, study_name: str) -> None:
    r = resources.StudyResource.from_name(study_name)
    with self._lock:
      try:
        del self._owners[r.owner_id][r.study_id]
      except KeyError:
        raise custom_errors.NotFoundError(f'Study not found: {study_name}')

  def list_studies(self, owner_name: str) -> List:
    r = resources.OwnerResource.from_name(owner_name)
    with self._lock:
      if not self._owners.get(r.owner_id):
        # Unknown owner, or an owner whose studies were all deleted: both
        # raise, matching the SQL implementation's row-based behavior.
        raise custom_errors.NotFoundError(f'Owner not found: {owner_name}')
      return [copy.deepcopy(node.study_proto)
              for node in self._owners[r.owner_id].values()]

  # -- trials ---------------------------------------------------------------

  def create_trial(self, trial) -> str:
    r = resources.TrialResource.from_name(trial.name)
    with self._lock:
      node = self._study_node(r.study_resource.name)
      if r.trial_id in node.trials:
        raise custom_errors.AlreadyExistsError(
            f'Trial already exists: {trial.name}')
      node.trials[r.trial_id] = copy.deepcopy(trial)
    return trial.name

  def get_trial(self, trial_name: str):
    r = resources.TrialResource.from_name(trial_name)
    with self._lock:
      node = self._study_node(r.study_resource.name)
      try:
        return copy.deepcopy(node.trials[r.trial_id])
      except KeyError:
        raise custom_errors.NotFoundError(f'Trial not found: {trial_name}')

  def update_trial(self, trial) -> str:
    r = resources.TrialResource.from_name(trial.name)
    with self._lock:
      node = self._study_node(r.study_resource.name)
      if r.trial_id not in node.trials:
        raise custom_errors.NotFoundError(f'Trial not found: {trial.name}')
      node.trials[r.trial_id] = copy.deepcopy(trial)
    return trial.name

  def list_trials(self, study_name: str) -> List:
    with self._lock:
      node = self._study_node(study_name)
      return [copy.deepcopy(t) for _, t in sorted(node.trials.items())]

  def delete_trial(self, trial_name: str) -> None:
    r = resources.TrialResource.from_name(trial_name)
    with self._lock:
      node = self._study_node(r.study_resource.name)
      if r.trial_id not in node.trials:
        raise custom_errors.NotFoundError(f'Trial not found: {trial_name}')
      del node.trials[r.trial_id]

  def max_trial_id(self, study_name: str) -> int:
    with self._lock:
      node = self._study_node(study_name)
      return max(node.trials.keys(), default=0)

  # -- suggestion operations ------------------------------------------------

  def create_suggestion_operation(self, operation) -> str:
    r = resources.SuggestionOperationResource.from_name(operation.name)
    study_name = resources.StudyResource(r.owner_id, r.study_id).name
    with self._lock:
      node = self._study_node(study_name)
      ops = node.suggestion_ops.setdefault(r.client_id, {})
      if r.operation_number in ops:
        raise custom_errors.AlreadyExistsError(
            f'Operation already exists: {operation.name}')
      ops[r.operation_number] = copy.deepcopy(operation)
    return operation.name

  def get_suggestion_operation(self, operation_name: str):
    r = resources.SuggestionOperationResource.from_name(operation_name)
    study_name = resources.StudyResource(r.owner_id, r.study_id).name
    with self._lock:
      node = self._study_node(study_name)
      try:
        return copy.deepcopy(
            node.suggestion_ops[r.client_id][r.operation_number])
      except KeyError:
        raise custom_errors.NotFoundError(
            f'Operation not found: {operation_name}')

  def update_suggestion_operation(self, operation) -> str:
    r = resources.SuggestionOperationResource.from_name(operation.name)
    study_name = resources.StudyResource(r.owner_id, r.study_id).name
    with self._lock:
      node = self._study_node(study_name)
      ops = node.suggestion_ops.get(r.client_id)
      if ops is None or r.operation_number not in ops:
        raise custom_errors.NotFoundError(
            f'Operation not found: {operation.name}')
      ops[r.operation_number] = copy.deepcopy(operation)
    return operation.name

  def list_suggestion_operations(
      self, study_name: str, client_id: str,
      filter_fn: Optional[Callable[[object], bool]] = None) -> List:
    with self._lock:
      node = self._study_node(study_name)
      if client_id not in node.suggestion_ops:
        raise custom_errors.NotFoundError(
            f'No operations for client: {client_id}')
      ops = [copy.deepcopy(op) for _, op in
             sorted(node.suggestion_ops[client_id].items())]
    if filter_fn is not None:
      ops = [op for op in ops if filter_fn(op)]
    return ops

  def max_suggestion_operation_number(self, study_name: str,
                                      client_id: str) -> int:
    with self._lock:
      node = self._study_node(study_name)
      if client_id not in node.suggestion_ops:
        raise custom_errors.NotFoundError(
            f'No operations for client: {client_id}')
      return max(node.suggestion_ops[client_id].keys(), default=0)

  # -- early-stopping operations -------------------------------------------

  def create_early_stopping_operation(self, operation) -> str:
    r = resources.EarlyStoppingOperationResource.from_name(operation.name)
    study_name = resources.StudyResource(r.owner_id, r.study_id).name
    with self._lock:
      node = self._study_node(study_name)
      if r.operation_id in node.early_stopping_ops:
        raise custom_errors.AlreadyExistsError(
            f'Operation already exists: {operation.name}')
      node.early_stopping_ops[r.operation_id] = copy.deepcopy(operation)
    return operation.name

  def get_early_stopping_operation(self, operation_name: str):
    r = resources.EarlyStoppingOperationResource.from_name(operation_name)
    study_name = resources.StudyResource(r.owner_id, r.study_id).name
    with self._lock:
      node = self._study_node(study_name)
      try:
        return copy.deepcopy(node.early_stopping_ops[r.operation_id])
      except KeyError:
        raise custom_errors.NotFoundError(
            f'Operation not found: {operation_name}')

  def update_early_stopping_operation(self, operation) -> str:
    r = resources.EarlyStoppingOperationResource.from_name(operation.name)
    study_name = resources.StudyResource(r.owner_id, r.study_id).name
    with self._lock:
      node = self._study_node(study_name)
      if r.operation_id not in node.early_stopping_ops:
        raise custom_errors.NotFoundError(
            f'Operation not found: {operation.name}')
      node.early_stopping_ops[r.operation_id] = copy.deepcopy(operation)
    return operation.name

  # -- metadata -------------------------------------------------------------

  def update_metadata(self, study_name: str, study_metadata: Iterable,
                      trial_metadata: Iterable) -> None:
    r = resources.StudyResource.from_name(study_name)
    with self._lock:
      node = self._study_node(study_name)
      metadata_util.merge_study_metadata(node.study_proto.study_spec,
                                         study_metadata)
      for update in trial_metadata:
        trial_id = int(update.trial_id)
        if trial_id not in node.trials:
          raise KeyError(f'Trial {trial_id} not found in {study_name}')
        metadata_util.merge_trial_metadata(node.trials[trial_id],
                                           [update.metadatum])
