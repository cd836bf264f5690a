"""Service-level pyvizier API (parity with vizier/service/pyvizier).

The shared data model plus OSS service extensions (StudyConfig,
proto converters, pythia study descriptors).
"""

from vizier_amd._src.pyvizier import metadata_util
from vizier_amd._src.pyvizier.automated_stopping import (
    AutomatedStoppingConfig,
    AutomatedStoppingConfigProto,
)
from vizier_amd._src.pyvizier.base_study_config import (
    MetricInformation,
    MetricsConfig,
    ObjectiveMetricGoal,
    ProblemStatement,
)
from vizier_amd._src.pyvizier.common import (
    Metadata,
    MetadataValue,
    Namespace,
)
from vizier_amd._src.pyvizier.parameter_config import (
    ExternalType,
    ParameterConfig,
    SearchSpace,
    SearchSpaceSelector,
)
from vizier_amd._src.pyvizier.proto_converters import (
    MeasurementConverter,
    MetadataDeltaConverter,
    MonotypeParameterSequence,
    ParameterConfigConverter,
    ParameterType,
    ParameterValueConverter,
    ProblemStatementConverter,
    ScaleType,
    StudyStateConverter,
    TrialConverter,
    TrialSuggestionConverter,
)
from vizier_amd._src.pythia.policy import (
    StudyDescriptor,
    StudyState,
    StudyStateInfo,
)
from vizier_amd._src.pyvizier.study import ProblemAndTrials
from vizier_amd._src.pyvizier.study_config import (
    Algorithm,
    ObservationNoise,
    StudyConfig,
)
from vizier_amd._src.pyvizier.trial import (
    CompletedTrial,
    CompletedTrialWithMeasurements,
    Measurement,
    MetadataDelta,
    Metric,
    NaNMetric,
    ParameterDict,
    ParameterValue,
    PendingTrial,
    PendingTrialWithMeasurements,
    Trial,
    TrialFilter,
    TrialStatus,
    TrialSuggestion,
)
from vizier_amd.pyvizier import ParameterValueSequence
from vizier_amd._src.service.pythia_converters import (
    EarlyStopConverter,
    SuggestConverter,
)
