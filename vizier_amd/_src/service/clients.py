"""High-level user-facing clients.

Capability parity with vizier/_src/service/clients.py (Trial :39,
Study :126 with from_study_config :236).
"""

from __future__ import annotations

from typing import Any, Callable, Iterator, List, Mapping, Optional, Type

from vizier_amd import pyvizier as vz
from vizier_amd.client import client_abc
from vizier_amd.client.client_abc import ResourceNotFoundError
from vizier_amd._src.service.constants import UNUSED_CLIENT_ID
from vizier_amd._src.pyvizier import proto_converters as pc
from vizier_amd._src.service import resources
from vizier_amd._src.service import vizier_client
from vizier_amd._src.service.proto import study_pb2

# Re-exported so users can tweak polling/endpoint like the reference.
environment_variables = vizier_client.environment_variables

_STUDY_STATE_TO_PROTO = {
    vz.StudyState.ACTIVE: study_pb2.Study.State.Value('ACTIVE'),
    vz.StudyState.ABORTED: study_pb2.Study.State.Value('INACTIVE'),
    vz.StudyState.COMPLETED: study_pb2.Study.State.Value('COMPLETED'),
}
_STUDY_STATE_FROM_PROTO = {
    study_pb2.Study.State.Value('STATE_UNSPECIFIED'): vz.StudyState.ACTIVE,
    study_pb2.Study.State.Value('ACTIVE'): vz.StudyState.ACTIVE,
    study_pb2.Study.State.Value('INACTIVE'): vz.StudyState.ABORTED,
    study_pb2.Study.State.Value('COMPLETED'): vz.StudyState.COMPLETED,
}


class Trial(client_abc.TrialInterface):
  """Client handle to one trial."""

  def __init__(self, client: vizier_client.VizierClient, uid: int):
    self._client = client
    self._id = uid

  @property
  def id(self) -> int:
    return self._id

  @property
  def parameters(self) -> Mapping[str, Any]:
    trial = self.materialize(include_all_measurements=False)
    study_config = self._client.get_study_config()
    return study_config.pytrial_parameters(trial)

  def delete(self) -> None:
    self._client.delete_trial(self._id)

  def update_metadata(self, delta: vz.Metadata) -> None:
    md_delta = vz.MetadataDelta(on_trials={self._id: delta})
    self._client.update_metadata(md_delta)

  def complete(self, measurement: Optional[vz.Measurement] = None, *,
               infeasible_reason: Optional[str] = None
               ) -> Optional[vz.Measurement]:
    trial = self._client.complete_trial(self._id, measurement,
                                        infeasible_reason)
    return trial.final_measurement

  def check_early_stopping(self) -> bool:
    return self._client.should_trial_stop(self._id)

  def stop(self) -> None:
    self._client.stop_trial(self._id)

  def add_measurement(self, measurement: vz.Measurement) -> None:
    self._client.report_intermediate_objective_value(
        int(measurement.steps), measurement.elapsed_secs,
        [measurement.as_float_dict()], self._id)

  def materialize(self, *, include_all_measurements: bool = True) -> vz.Trial:
    trial = self._client.get_trial(self._id)
    if not include_all_measurements:
      trial.measurements.clear()
    return trial

  @property
  def study(self) -> 'Study':
    return Study(self._client)


class TrialIterable(client_abc.TrialIterable):

  def __init__(self, trials: List[vz.Trial],
               client: vizier_client.VizierClient):
    self._trials = trials
    self._client = client

  def __iter__(self) -> Iterator[Trial]:
    for t in self._trials:
      yield Trial(self._client, t.id)

  def get(self) -> Iterator[vz.Trial]:
    return iter(self._trials)

  def __len__(self) -> int:
    return len(self._trials)


class Study(client_abc.StudyInterface):
  """Client handle to one study."""

  def __init__(self, client: vizier_client.VizierClient):
    self._client = client

  @property
  def resource_name(self) -> str:
    return self._client.study_resource_name

  def suggest(self, *, count: Optional[int] = None,
              client_id: str = 'default_client_id') -> List[Trial]:
    trials = self._client.get_suggestions(count or 1,
                                          client_id_override=client_id)
    return [Trial(self._client, t.id) for t in trials]

  def delete(self) -> None:
    self._client.delete_study()

  def update_metadata(self, delta: vz.Metadata) -> None:
    self._client.update_metadata(vz.MetadataDelta(on_study=delta))

  def add_trial(self, trial: vz.Trial) -> Trial:
    added = self._client.add_trial(trial)
    return Trial(self._client, added.id)

  def request(self, suggestion: vz.TrialSuggestion) -> Trial:
    return self.add_trial(suggestion.to_trial(0))

  def trials(self, trial_filter: Optional[vz.TrialFilter] = None
             ) -> TrialIterable:
    all_trials = self._client.list_trials()
    if trial_filter is not None:
      all_trials = [t for t in all_trials if trial_filter(t)]
    return TrialIterable(all_trials, self._client)

  def get_trial(self, uid: int) -> Trial:
    try:
      trial = self._client.get_trial(uid)
      return Trial(self._client, trial.id)
    except Exception as e:
      raise client_abc.ResourceNotFoundError(
          f'Study f{self.resource_name} does not have trial {uid}.') from e

  def optimal_trials(self, *, count: Optional[int] = None) -> TrialIterable:
    trials = self._client.list_optimal_trials()
    if count is not None:
      trials = trials[:count]
    return TrialIterable(trials, self._client)

  def materialize_problem_statement(self) -> vz.ProblemStatement:
    return self.materialize_study_config().to_problem()

  def materialize_study_config(self) -> vz.StudyConfig:
    return self._client.get_study_config()

  def set_state(self, state: vz.StudyState) -> None:
    self._client.set_study_state(_STUDY_STATE_TO_PROTO[state])

  def materialize_state(self) -> vz.StudyState:
    return _STUDY_STATE_FROM_PROTO[self._client.get_study_state()]

  @classmethod
  def from_resource_name(cls, name: str) -> 'Study':
    client = vizier_client.VizierClient(name, 'default_client_id')
    return cls(client)

  @classmethod
  def from_owner_and_id(cls, owner_id: str, study_id: str) -> 'Study':
    return cls.from_resource_name(
        resources.StudyResource(owner_id, study_id).name)

  @classmethod
  def from_study_config(cls, config: vz.StudyConfig, *, owner: str,
                        study_id: str) -> 'Study':
    client = vizier_client.create_or_load_study(
        owner, 'default_client_id', study_id, config)
    return cls(client)
