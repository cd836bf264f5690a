"""Public pyvizier API: the cross-platform data model plus OSS extensions.

Usage parity with `from vizier import pyvizier as vz` in the reference.
"""

from vizier_amd._src.pyvizier.automated_stopping import AutomatedStoppingConfig
from vizier_amd._src.pyvizier.base_study_config import (
    MetricInformation,
    MetricsConfig,
    MetricType,
    ObjectiveMetricGoal,
    ProblemStatement,
)
from vizier_amd._src.pyvizier.common import Metadata, MetadataValue, Namespace
from vizier_amd._src.pyvizier.context import Context
from vizier_amd._src.pyvizier.parameter_config import (
    ExternalType,
    FidelityConfig,
    FidelityMode,
    InvalidParameterError,
    MonotypeParameterSequence,
    ParameterConfig,
    ParameterConfigSelector,
    ParameterType,
    ParameterValueTypes,
    ScaleType,
    SearchSpace,
    SearchSpaceSelector,
)
from vizier_amd._src.pyvizier.study import ProblemAndTrials
from vizier_amd._src.pyvizier.study_config import (
    Algorithm,
    ObservationNoise,
    StudyConfig,
)
from vizier_amd._src.pyvizier.trial import (
    CompletedTrial,
    Measurement,
    MetadataDelta,
    Metric,
    NaNMetric,
    ParameterDict,
    ParameterValue,
    PendingTrial,
    Trial,
    TrialFilter,
    TrialStatus,
    TrialSuggestion,
)

# State enums used by clients (mirrors pyvizier.StudyState).
import enum as _enum


class StudyState(_enum.Enum):
  ACTIVE = 'ACTIVE'
  ABORTED = 'ABORTED'
  COMPLETED = 'COMPLETED'


class StudyStateInfo:
  def __init__(self, state: StudyState, details: str = ''):
    self.state = StudyState(state)
    self.details = details

# Reference-compat additions (vizier/pyvizier/__init__.py parity).
from typing import Sequence as _Sequence, Union as _Union

from vizier_amd._src.pyvizier.parameter_iterators import (
    SequentialParameterBuilder,
)
from vizier_amd._src.pyvizier.trial import (
    CompletedTrialWithMeasurements,
    PendingTrialWithMeasurements,
)
from vizier_amd._src.pythia.policy import StudyDescriptor

ParameterValueSequence = _Union[ParameterValueTypes, _Sequence[int],
                                _Sequence[float], _Sequence[str],
                                _Sequence[bool]]
