"""The Pythia policy protocol: requests/decisions + the Policy ABC.

Capability parity with vizier/_src/pythia/policy.py (SuggestRequest :41,
SuggestDecision, EarlyStopRequest/Decisions :*, Policy :207-266).
"""

from __future__ import annotations

import abc
import dataclasses
import enum
from typing import FrozenSet, Iterable, List, Optional, Sequence, Union

from vizier_amd._src.pyvizier.base_study_config import ProblemStatement
from vizier_amd._src.pyvizier.common import Metadata
from vizier_amd._src.pyvizier.study_config import StudyConfig
from vizier_amd._src.pyvizier.trial import (
    Measurement,
    MetadataDelta,
    TrialSuggestion,
)


class StudyState(enum.Enum):
  """Cross-platform study lifecycle state (pythia/study.py:25)."""
  ACTIVE = 'ACTIVE'
  ABORTED = 'ABORTED'
  COMPLETED = 'COMPLETED'


@dataclasses.dataclass
class StudyStateInfo:
  state: StudyState
  explanation: str = ''

  def __post_init__(self):
    self.state = StudyState(self.state)


@dataclasses.dataclass(frozen=True)
class StudyDescriptor:
  """Identifies a study and carries its configuration."""

  config: Union[StudyConfig, ProblemStatement]
  guid: str = ''
  max_trial_id: int = 0


@dataclasses.dataclass(frozen=True)
class SuggestRequest:
  """Asks a policy for `count` suggestions."""

  study_descriptor: StudyDescriptor
  count: int = 1
  checkpoint_dir: Optional[str] = None

  @property
  def study_config(self):
    return self.study_descriptor.config

  @property
  def study_guid(self) -> str:
    return self.study_descriptor.guid

  @property
  def max_trial_id(self) -> int:
    return self.study_descriptor.max_trial_id


class SuggestDecision:
  """Suggestions plus metadata changes."""

  def __init__(self, suggestions: Iterable[TrialSuggestion],
               metadata: Optional[MetadataDelta] = None):
    self.suggestions: List[TrialSuggestion] = list(suggestions)
    self.metadata = metadata if metadata is not None else MetadataDelta()

  def __len__(self) -> int:
    return len(self.suggestions)


@dataclasses.dataclass(frozen=True)
class EarlyStopRequest:
  """Asks a policy which trials should stop."""

  study_descriptor: StudyDescriptor
  trial_ids: FrozenSet[int] = frozenset()
  checkpoint_dir: Optional[str] = None

  def __post_init__(self):
    object.__setattr__(self, 'trial_ids', frozenset(self.trial_ids))

  @property
  def study_config(self):
    return self.study_descriptor.config

  @property
  def study_guid(self) -> str:
    return self.study_descriptor.guid

  @property
  def max_trial_id(self) -> int:
    return self.study_descriptor.max_trial_id


@dataclasses.dataclass
class EarlyStopDecision:
  """Stop/continue decision for one trial."""

  id: int
  reason: str = ''
  should_stop: bool = True
  metadata: Metadata = dataclasses.field(default_factory=Metadata)
  predicted_final_measurement: Optional[Measurement] = None


class EarlyStopDecisions:
  """A batch of early-stopping decisions plus metadata changes."""

  def __init__(self, decisions: Iterable[EarlyStopDecision] = (),
               metadata: Optional[MetadataDelta] = None):
    self.decisions: List[EarlyStopDecision] = list(decisions)
    self.metadata = metadata if metadata is not None else MetadataDelta()


class Policy(abc.ABC):
  """The algorithm-side interface of the Pythia protocol."""

  @abc.abstractmethod
  def suggest(self, request: SuggestRequest) -> SuggestDecision:
    """Returns suggestions (and optional metadata updates)."""

  def early_stop(self, request: EarlyStopRequest) -> EarlyStopDecisions:
    """Returns early-stopping decisions; default: never stop."""
    decisions = [EarlyStopDecision(id=tid, reason='Default: do not stop.',
                                   should_stop=False)
                 for tid in request.trial_ids]
    return EarlyStopDecisions(decisions=decisions)

  @property
  def should_be_cached(self) -> bool:
    """True if the service should keep this policy instance alive."""
    return False
