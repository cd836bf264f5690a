"""PyGlove integration: converters, policy bridge, backend init.

Capability parity with vizier/_src/pyglove/ (converters.VizierConverter,
pythia.create_policy :176, oss_vizier.init :264, algorithms
.BuiltinAlgorithm :64). PyGlove is an optional dependency (absent from
this image): every entry point imports it lazily and raises a clear
ImportError otherwise; `vizier.pyglove` exposes these via a module
__getattr__ so the facade imports cleanly without pyglove.
"""

from __future__ import annotations

from typing import Optional, Sequence

from vizier_amd import pyvizier as vz
from vizier_amd._src.pythia.policy import (
    Policy,
    SuggestDecision,
    SuggestRequest,
)


def _pg():
  try:
    import pyglove as pg  # noqa: F401
    return pg
  except ImportError as e:  # pragma: no cover - pyglove not in image
    raise ImportError(
        'The pyglove integration requires the `pyglove` package '
        '(not installed in this environment).') from e


class VizierConverter:
  """vz.SearchSpace/Trials <-> pyglove DNA (flat spaces)."""

  def __init__(self, problem: vz.ProblemStatement):
    self._problem = problem
    pg = _pg()
    specs = []
    for p in problem.search_space.parameters:
      if p.type == vz.ParameterType.DOUBLE:
        lo, hi = p.bounds
        specs.append(pg.floatv(lo, hi, name=p.name))
      elif p.type == vz.ParameterType.INTEGER:
        lo, hi = p.bounds
        specs.append(pg.oneof(list(range(int(lo), int(hi) + 1)),
                              name=p.name))
      elif p.type in (vz.ParameterType.CATEGORICAL,
                      vz.ParameterType.DISCRETE):
        specs.append(pg.oneof(list(p.feasible_values), name=p.name))
      else:
        raise ValueError(f'Unsupported parameter: {p}')
    self._space = pg.Dict(
        {p.name: s for p, s in zip(problem.search_space.parameters,
                                   specs)})
    self.dna_spec = pg.dna_spec(self._space)
    self.metrics_to_optimize = [
        m.name for m in problem.metric_information]

  @classmethod
  def from_problem(cls, problem: vz.ProblemStatement
                   ) -> 'VizierConverter':
    return cls(problem)

  def to_parameters(self, dna) -> vz.ParameterDict:
    pg = _pg()
    value = pg.materialize(self._space, dna)
    return vz.ParameterDict({k: v for k, v in value.items()})

  def to_dna(self, trial: vz.Trial):
    pg = _pg()
    values = {name: trial.parameters.get_value(name)
              for name in (p.name for p in
                           self._problem.search_space.parameters)}
    return pg.DNA.from_dict(values, self.dna_spec, use_ints_as_literals=True)

  def reward_of(self, trial: vz.Trial) -> float:
    metric = self._problem.metric_information.item()
    value = trial.final_measurement.metrics[metric.name].value
    return value if metric.goal.is_maximize else -value


class PyGlovePolicy(Policy):
  """Pythia policy driven by a pyglove DNAGenerator."""

  def __init__(self, supporter, problem: vz.ProblemStatement, algorithm,
               prior_trials: Optional[Sequence[vz.Trial]] = None):
    self._supporter = supporter
    self._problem = problem
    self._converter = VizierConverter.from_problem(problem)
    self._algorithm = algorithm
    algorithm.setup(self._converter.dna_spec)
    for trial in prior_trials or ():
      if trial.final_measurement is not None:
        dna = self._converter.to_dna(trial)
        self._algorithm.feedback(dna, self._converter.reward_of(trial))
    self._fed_ids = set()

  def suggest(self, request: SuggestRequest) -> SuggestDecision:
    completed = self._supporter.GetTrials(
        study_guid=request.study_guid,
        status_matches=vz.TrialStatus.COMPLETED)
    for trial in completed:
      if trial.id in self._fed_ids:
        continue
      self._fed_ids.add(trial.id)
      self._algorithm.feedback(self._converter.to_dna(trial),
                               self._converter.reward_of(trial))
    suggestions = []
    for _ in range(request.count or 1):
      dna = self._algorithm.propose()
      suggestions.append(
          vz.TrialSuggestion(self._converter.to_parameters(dna)))
    return SuggestDecision(suggestions)


def create_policy(supporter, problem_statement: vz.ProblemStatement,
                  algorithm, early_stopping_policy=None,
                  prior_trials: Optional[Sequence[vz.Trial]] = None
                  ) -> Policy:
  """Creates a Pythia policy that runs a PyGlove algorithm."""
  del early_stopping_policy  # Early stopping not bridged yet.
  return PyGlovePolicy(supporter, problem_statement, algorithm,
                       prior_trials)


def make_builtin_algorithm_class():
  """Lazily defines BuiltinAlgorithm (a pg.DNAGenerator marker whose
  `name` selects a built-in Vizier algorithm string)."""
  pg = _pg()

  class BuiltinAlgorithm(pg.geno.DNAGenerator):
    """Marker: route suggestion to the named built-in algorithm."""

    def __init__(self, name: str = 'DEFAULT'):
      super().__init__()
      self._name = name

    @property
    def name(self) -> str:
      return self._name

    @property
    def multi_objective(self) -> bool:
      return self._name in ('DEFAULT', 'GP_BANDIT',
                            'LINEAR_COMBINATION_SEARCH', 'RANDOM_SEARCH')

    def _propose(self):
      raise NotImplementedError(
          'BuiltinAlgorithm is resolved by the Vizier backend, not '
          'sampled directly.')

  return BuiltinAlgorithm


_BACKEND = {}


def init(study_prefix: Optional[str] = None,
         vizier_endpoint: Optional[str] = None,
         pythia_port: Optional[int] = None) -> None:
  """Initializes the OSS Vizier backend for pg.sample.

  With no endpoint, an in-process Vizier service is used (the same
  NO_ENDPOINT path as vizier_client).
  """
  _pg()  # Validate pyglove availability eagerly, like the reference.
  from vizier_amd._src.service import clients as service_clients
  if vizier_endpoint is not None:
    service_clients.environment_variables.server_endpoint = \
        vizier_endpoint
  _BACKEND['study_prefix'] = study_prefix
  _BACKEND['pythia_port'] = pythia_port
