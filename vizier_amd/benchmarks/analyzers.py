"""Public analyzers API (parity with vizier/benchmarks/analyzers.py)."""

from vizier_amd._src.benchmarks.analyzers.convergence_curve import (
    ConvergenceCurve,
    ConvergenceCurveConverter,
    HypervolumeCurveConverter,
    LogEfficiencyConvergenceCurveComparator,
    PercentageBetterConvergenceCurveComparator,
    WinRateComparator,
)
from vizier_amd._src.benchmarks.analyzers.simple_regret_score import (
    t_test_less_mean_score,
    t_test_mean_score,
)
from vizier_amd._src.benchmarks.analyzers.state_analyzer import (
    BenchmarkRecord,
    BenchmarkStateAnalyzer,
    compute_parameter_entropy,
)

from vizier_amd._src.benchmarks.analyzers.convergence_curve import (
    ConvergenceComparator,
    ConvergenceComparatorFactory,
    LogEfficiencyConvergenceCurveComparatorFactory,
    MultiMetricCurveConverter,
    PercentageBetterConvergenceCurveComparatorFactory,
    RestartingCurveConverter,
    StatefulCurveConverter,
    WinRateConvergenceCurveComparator,
    WinRateConvergenceCurveComparatorFactory,
)
from vizier_amd._src.benchmarks.analyzers.plot_utils import (
    plot_from_records,
    plot_mean_convergence,
    plot_median_convergence,
)
from vizier_amd._src.benchmarks.analyzers.state_analyzer import (
    BenchmarkRecordAnalyzer,
    PlotElement,
)
