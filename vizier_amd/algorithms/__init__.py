"""Public algorithms API (parity with vizier/algorithms)."""

from vizier_amd._src.algorithms.core.abstractions import (
    ActiveTrials,
    CompletedTrials,
    Designer,
    PartiallySerializableDesigner,
    Prediction,
    Predictor,
    SerializableDesigner,
)
from vizier_amd._src.algorithms.policies.designer_policy import (
    DesignerPolicy,
    PartiallySerializableDesignerPolicy,
)
from vizier_amd._src.algorithms.policies.random_policy import RandomPolicy
from vizier_amd._src.pythia.local_policy_supporters import (
    InRamPolicySupporter,
)

from vizier_amd._src.algorithms.core.abstractions import DesignerFactory
from vizier_amd._src.algorithms.optimizers.base import (
    BatchTrialScoreFunction,
    BranchSelection,
    BranchSelector,
    BranchThenOptimizer,
    GradientFreeOptimizer,
)
from vizier_amd._src.algorithms.policies.designer_policy import (
    InRamDesignerPolicy,
    SerializableDesignerPolicy,
)
