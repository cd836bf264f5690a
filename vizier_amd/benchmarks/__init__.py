"""Public benchmarks API (parity with vizier/benchmarks)."""

from vizier_amd._src.benchmarks.experimenters.experimenter import (
    Experimenter,
)
from vizier_amd._src.benchmarks.experimenters.experimenter_factory import (
    BBOBExperimenterFactory,
    SingleObjectiveExperimenterFactory,
)
from vizier_amd._src.benchmarks.experimenters.numpy_experimenter import (
    NumpyExperimenter,
)
from vizier_amd._src.benchmarks.experimenters.wrappers import (
    DiscretizingExperimenter,
    InfeasibleExperimenter,
    MultiObjectiveExperimenter,
    NoisyExperimenter,
    NormalizingExperimenter,
    ShiftingExperimenter,
    SignFlipExperimenter,
    SwitchExperimenter,
)
from vizier_amd._src.benchmarks.runners.benchmark_runner import (
    BenchmarkRunner,
    BenchmarkSubroutine,
    EvaluateActiveTrials,
    GenerateAndEvaluate,
    GenerateSuggestions,
)
from vizier_amd._src.benchmarks.runners.benchmark_state import (
    BenchmarkState,
    DesignerBenchmarkStateFactory,
    PolicySuggester,
)
from vizier_amd._src.benchmarks.analyzers.convergence_curve import (
    ConvergenceCurve,
    ConvergenceCurveConverter,
    HypervolumeCurveConverter,
    LogEfficiencyConvergenceCurveComparator,
    PercentageBetterConvergenceCurveComparator,
    WinRateComparator,
)

from vizier_amd._src.benchmarks.runners.benchmark_runner import (
    FillActiveTrials,
)
from vizier_amd._src.benchmarks.runners.benchmark_state import (
    BenchmarkStateFactory,
    ExperimenterDesignerBenchmarkStateFactory,
    PolicyBenchmarkStateFactory,
    SeededPolicyFactory,
)
