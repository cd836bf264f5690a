"""PolicySupporter backed by the Vizier service.

Capability parity with vizier/_src/service/service_policy_supporter.py:32-95.
"""

from __future__ import annotations

from typing import Iterable, List, Optional

from vizier_amd._src.pyvizier import proto_converters as pc
from vizier_amd._src.pyvizier.study_config import StudyConfig
from vizier_amd._src.pyvizier.trial import Trial, TrialStatus
from vizier_amd._src.pythia.policy_supporter import PolicySupporter
from vizier_amd._src.service.proto import vizier_service_pb2


class ServicePolicySupporter(PolicySupporter):
  """Fetches study/trial data through a VizierService."""

  def __init__(self, study_guid: str, vizier_service):
    self._study_guid = study_guid
    self._vizier_service = vizier_service

  def GetStudyConfig(self, study_guid: Optional[str] = None) -> StudyConfig:
    study_guid = study_guid or self._study_guid
    study = self._vizier_service.GetStudy(
        vizier_service_pb2.GetStudyRequest(name=study_guid))
    return StudyConfig.from_proto(study.study_spec)

  def GetTrials(self, *, study_guid: Optional[str] = None,
                trial_ids: Optional[Iterable[int]] = None,
                min_trial_id: Optional[int] = None,
                max_trial_id: Optional[int] = None,
                status_matches: Optional[TrialStatus] = None,
                include_intermediate_measurements: bool = True
                ) -> List[Trial]:
    study_guid = study_guid or self._study_guid
    resp = self._vizier_service.ListTrials(
        vizier_service_pb2.ListTrialsRequest(parent=study_guid))
    trials = pc.TrialConverter.from_protos(resp.trials)
    ids = frozenset(trial_ids) if trial_ids is not None else None
    out = []
    for t in trials:
      if ids is not None and t.id not in ids:
        continue
      if min_trial_id is not None and t.id < min_trial_id:
        continue
      if max_trial_id is not None and t.id > max_trial_id:
        continue
      if status_matches is not None and t.status != status_matches:
        continue
      out.append(t)
    return out
