"""Public Pythia API (parity with `from vizier import pythia`)."""

from vizier_amd._src.pythia.local_policy_supporters import (
    InRamPolicySupporter,
)
from vizier_amd._src.pythia.policy import (
    EarlyStopDecision,
    EarlyStopDecisions,
    EarlyStopRequest,
    Policy,
    StudyDescriptor,
    SuggestDecision,
    SuggestRequest,
)
from vizier_amd._src.pythia.policy_factory import PolicyFactory
from vizier_amd._src.pythia.policy_supporter import PolicySupporter
from vizier_amd._src.pythia.pythia_errors import (
    CachedPolicyIsStaleError,
    InactivateStudyError,
    LoadTooLargeError,
    PythiaError,
    PythiaProtocolError,
    TemporaryPythiaError,
    VizierDatabaseError,
)
