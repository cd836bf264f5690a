"""Data access interface policies use to read the study.

Capability parity with vizier/_src/pythia/policy_supporter.py:26-133.
"""

from __future__ import annotations

import abc
import datetime
from typing import Iterable, List, Optional

from vizier_amd._src.pyvizier.study_config import StudyConfig
from vizier_amd._src.pyvizier.trial import Trial, TrialStatus
from vizier_amd._src.pythia import pythia_errors


class PolicySupporter(abc.ABC):
  """Used by Policies to communicate with Vizier."""

  @abc.abstractmethod
  def GetStudyConfig(self, study_guid: Optional[str] = None) -> StudyConfig:
    """Returns the study config for `study_guid` (default: current study)."""

  @abc.abstractmethod
  def GetTrials(self, *, study_guid: Optional[str] = None,
                trial_ids: Optional[Iterable[int]] = None,
                min_trial_id: Optional[int] = None,
                max_trial_id: Optional[int] = None,
                status_matches: Optional[TrialStatus] = None,
                include_intermediate_measurements: bool = True
                ) -> List[Trial]:
    """Returns trials matching the filter."""

  def CheckCancelled(self, note: Optional[str] = None) -> None:
    """Raises CancelComputeError if the RPC has been cancelled."""

  def TimeRemaining(self) -> datetime.timedelta:
    """Time remaining to compute a result (default: unbounded)."""
    return datetime.timedelta.max

  def SendMetadata(self, delta) -> None:
    """Immediately writes a MetadataDelta to the backing store.

    Optional: policies usually return metadata in their decision instead.
    """
    raise NotImplementedError
