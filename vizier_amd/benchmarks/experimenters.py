"""Public experimenters API (parity with vizier/benchmarks/experimenters)."""

from vizier_amd._src.benchmarks.experimenters import experimenter_factory
from vizier_amd._src.benchmarks.experimenters.combo import (
    CentroidExperimenter,
    ContaminationExperimenter,
    IsingExperimenter,
    MAXSATExperimenter,
    PestControlExperimenter,
)
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
from vizier_amd._src.benchmarks.experimenters.synthetic import bbob

from vizier_amd._src.benchmarks.experimenters.extra import (
    CombinedExperimenterFactory,
    ExperimenterFactory,
    HashingInfeasibleExperimenter,
    HyperCubeExperimenter,
    L1CategorialExperimenter,
    MultiObjectiveNumpyExperimenter,
    ParamRegionInfeasibleExperimenter,
    PredictorExperimenter,
    SerializableExperimenterFactory,
)
from vizier_amd._src.benchmarks.experimenters.synthetic.classic import (
    BernoulliMultiArmExperimenter,
    Branin2DExperimenter,
    DHExperimenter,
    FixedMultiArmExperimenter,
    HartmannExperimenter,
)
from vizier_amd._src.benchmarks.experimenters.synthetic.mo_problems import (
    DTLZExperimenterFactory,
    WFGExperimenterFactory,
    ZDTExperimenterFactory,
)
from vizier_amd._src.benchmarks.experimenters.synthetic.simplekd import (
    SimpleKDExperimenter,
)
from vizier_amd._src.benchmarks.experimenters.wrappers import (
    PermutingExperimenter,
    SparseExperimenter,
)
