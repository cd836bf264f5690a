"""Composable experimenter decorators.

Capability parity with the wrapper experimenters under
vizier/_src/benchmarks/experimenters/ (ShiftingExperimenter,
NoisyExperimenter, DiscretizingExperimenter, NormalizingExperimenter,
SignFlipExperimenter, InfeasibleExperimenter, SwitchExperimenter,
MultiObjectiveExperimenter).
"""

from __future__ import annotations

import copy
from typing import Callable, Dict, Optional, Sequence

import numpy as np

from vizier_amd import pyvizier as vz
from vizier_amd._src.benchmarks.experimenters.experimenter import (
    Experimenter,
)


class ShiftingExperimenter(Experimenter):
  """Evaluates f(x + shift); bounds are narrowed so x+shift stays valid."""

  def __init__(self, exptr: Experimenter, shift: np.ndarray):
    self._exptr = exptr
    self._shift = np.asarray(shift, dtype=np.float64)
    self._problem = exptr.problem_statement()
    params = self._problem.search_space.parameters
    if len(self._shift) != len(params):
      raise ValueError('Shift dimension mismatch.')
    self._names = [pc.name for pc in params]

  def evaluate(self, suggestions: Sequence[vz.Trial]) -> None:
    shifted = []
    for trial in suggestions:
      t = copy.deepcopy(trial)
      for name, s in zip(self._names, self._shift):
        t.parameters[name] = float(t.parameters.get_value(name)) + s
      shifted.append(t)
    self._exptr.evaluate(shifted)
    for trial, t in zip(suggestions, shifted):
      trial.complete(t.final_measurement,
                     infeasibility_reason=t.infeasibility_reason)

  def problem_statement(self) -> vz.ProblemStatement:
    return copy.deepcopy(self._problem)


class NoisyExperimenter(Experimenter):
  """Adds observation noise to every metric."""

  def __init__(self, exptr: Experimenter,
               noise_fn: Optional[Callable[[float], float]] = None, *,
               noise_std: float = 1.0, seed: Optional[int] = None):
    self._exptr = exptr
    rng = np.random.default_rng(seed)
    self._noise_fn = noise_fn or (
        lambda v: v + noise_std * rng.standard_normal())

  def evaluate(self, suggestions: Sequence[vz.Trial]) -> None:
    self._exptr.evaluate(suggestions)
    for trial in suggestions:
      if trial.final_measurement is None:
        continue
      for name, metric in list(trial.final_measurement.metrics.items()):
        trial.final_measurement.metrics[name] = self._noise_fn(
            metric.value)

  def problem_statement(self) -> vz.ProblemStatement:
    return self._exptr.problem_statement()


class DiscretizingExperimenter(Experimenter):
  """Exposes selected DOUBLE parameters as DISCRETE grids."""

  def __init__(self, exptr: Experimenter,
               discretization: Dict[str, Sequence[float]]):
    self._exptr = exptr
    self._discretization = {k: list(v) for k, v in discretization.items()}
    problem = exptr.problem_statement()
    space = vz.SearchSpace()
    for pc in problem.search_space.parameters:
      if pc.name in self._discretization:
        space.add(vz.ParameterConfig.factory(
            pc.name, feasible_values=self._discretization[pc.name]))
      else:
        space.add(pc)
    problem.search_space = space
    self._problem = problem

  def evaluate(self, suggestions: Sequence[vz.Trial]) -> None:
    self._exptr.evaluate(suggestions)

  def problem_statement(self) -> vz.ProblemStatement:
    return copy.deepcopy(self._problem)


class NormalizingExperimenter(Experimenter):
  """Normalizes metrics by |f| statistics estimated on a random grid."""

  def __init__(self, exptr: Experimenter, *, num_normalization_samples:
               int = 100, seed: Optional[int] = None):
    self._exptr = exptr
    problem = exptr.problem_statement()
    rng = np.random.default_rng(seed)
    from vizier_amd._src.algorithms.designers.random import (
        sample_parameters,
    )
    trials = []
    for i in range(num_normalization_samples):
      params = sample_parameters(rng, problem.search_space.parameters)
      trials.append(vz.Trial(params, id=i + 1))
    exptr.evaluate(trials)
    name = problem.metric_information.item().name
    values = [t.final_measurement.metrics[name].value for t in trials
              if t.final_measurement is not None]
    self._mean = float(np.mean(values))
    self._std = float(np.std(values)) or 1.0
    self._metric = name

  def evaluate(self, suggestions: Sequence[vz.Trial]) -> None:
    self._exptr.evaluate(suggestions)
    for trial in suggestions:
      if trial.final_measurement is None:
        continue
      v = trial.final_measurement.metrics[self._metric].value
      trial.final_measurement.metrics[self._metric] = \
          (v - self._mean) / self._std

  def problem_statement(self) -> vz.ProblemStatement:
    return self._exptr.problem_statement()


class SignFlipExperimenter(Experimenter):
  """Negates metrics (and flips MAXIMIZE <-> MINIMIZE)."""

  def __init__(self, exptr: Experimenter, *,
               flip_objective_goal: bool = True):
    self._exptr = exptr
    self._flip_goal = flip_objective_goal

  def evaluate(self, suggestions: Sequence[vz.Trial]) -> None:
    self._exptr.evaluate(suggestions)
    for trial in suggestions:
      if trial.final_measurement is None:
        continue
      for name, metric in list(trial.final_measurement.metrics.items()):
        trial.final_measurement.metrics[name] = -metric.value

  def problem_statement(self) -> vz.ProblemStatement:
    problem = self._exptr.problem_statement()
    if self._flip_goal:
      for mi in problem.metric_information:
        mi.flip_goal()
    return problem


class InfeasibleExperimenter(Experimenter):
  """Marks a random fraction of evaluations infeasible."""

  def __init__(self, exptr: Experimenter, infeasible_prob: float = 0.2,
               *, seed: Optional[int] = None):
    self._exptr = exptr
    self._prob = infeasible_prob
    self._rng = np.random.default_rng(seed)

  def evaluate(self, suggestions: Sequence[vz.Trial]) -> None:
    self._exptr.evaluate(suggestions)
    for trial in suggestions:
      if self._rng.random() < self._prob:
        trial.final_measurement = None
        trial.complete(vz.Measurement(),
                       infeasibility_reason='randomly infeasible')

  def problem_statement(self) -> vz.ProblemStatement:
    return self._exptr.problem_statement()


class SwitchExperimenter(Experimenter):
  """Routes evaluation through an extra INTEGER 'switch' parameter."""

  def __init__(self, experimenters: Sequence[Experimenter],
               switch_param: str = 'switch'):
    if not experimenters:
      raise ValueError('Need at least one experimenter.')
    self._exptrs = list(experimenters)
    self._switch = switch_param
    problem = self._exptrs[0].problem_statement()
    problem.search_space.root.add_int_param(
        switch_param, 0, len(self._exptrs) - 1)
    self._problem = problem

  def evaluate(self, suggestions: Sequence[vz.Trial]) -> None:
    for trial in suggestions:
      idx = int(trial.parameters.get_value(self._switch, 0))
      idx = min(max(idx, 0), len(self._exptrs) - 1)
      self._exptrs[idx].evaluate([trial])

  def problem_statement(self) -> vz.ProblemStatement:
    return copy.deepcopy(self._problem)


class MultiObjectiveExperimenter(Experimenter):
  """Combines single-objective experimenters over one search space."""

  def __init__(self, exptrs: Dict[str, Experimenter]):
    if not exptrs:
      raise ValueError('Need at least one experimenter.')
    self._exptrs = dict(exptrs)
    base = next(iter(self._exptrs.values())).problem_statement()
    metrics = []
    for name, e in self._exptrs.items():
      mi = e.problem_statement().metric_information.item()
      metrics.append(vz.MetricInformation(name=name, goal=mi.goal))
    self._problem = vz.ProblemStatement(
        search_space=base.search_space, metric_information=metrics)

  def evaluate(self, suggestions: Sequence[vz.Trial]) -> None:
    for trial in suggestions:
      combined: Dict[str, float] = {}
      for name, e in self._exptrs.items():
        t = copy.deepcopy(trial)
        t.final_measurement = None
        e.evaluate([t])
        inner = t.final_measurement.metrics
        combined[name] = next(iter(inner.values())).value
      trial.complete(vz.Measurement(metrics=combined))

  def problem_statement(self) -> vz.ProblemStatement:
    return copy.deepcopy(self._problem)


class PermutingExperimenter(Experimenter):
  """Permutes parameter values between named parameters before evaluating."""

  def __init__(self, exptr: Experimenter, parameters_to_permute:
               Sequence[str], *, seed: Optional[int] = None):
    self._exptr = exptr
    names = list(parameters_to_permute)
    rng = np.random.default_rng(seed)
    self._mapping = dict(zip(names,
                             [names[i] for i in rng.permutation(len(names))]))

  def evaluate(self, suggestions: Sequence[vz.Trial]) -> None:
    permuted = []
    for trial in suggestions:
      t = copy.deepcopy(trial)
      values = {src: t.parameters.get_value(src)
                for src in self._mapping}
      for src, dst in self._mapping.items():
        t.parameters[dst] = values[src]
      permuted.append(t)
    self._exptr.evaluate(permuted)
    for trial, t in zip(suggestions, permuted):
      if t.final_measurement is not None:
        trial.complete(t.final_measurement,
                       infeasibility_reason=t.infeasibility_reason)

  def problem_statement(self) -> vz.ProblemStatement:
    return self._exptr.problem_statement()


class SparseExperimenter(Experimenter):
  """Embeds the problem in a higher-dimensional space of inert parameters."""

  def __init__(self, exptr: Experimenter, num_dummy_dimensions: int,
               *, prefix: str = 'dummy'):
    self._exptr = exptr
    self._problem = exptr.problem_statement()
    for i in range(num_dummy_dimensions):
      self._problem.search_space.root.add_float_param(
          f'{prefix}{i}', 0.0, 1.0)

  def evaluate(self, suggestions: Sequence[vz.Trial]) -> None:
    self._exptr.evaluate(suggestions)

  def problem_statement(self) -> vz.ProblemStatement:
    return copy.deepcopy(self._problem)


class SurrogateExperimenter(Experimenter):
  """Uses a trained regression model as the objective (HPO-B style).

  The reference's HPO-B / NAS-Bench handlers evaluate offline-trained
  surrogates; this wrapper provides the same mechanism for any
  sklearn-style `.predict(features)` model over the converter features.
  """

  def __init__(self, model, problem: vz.ProblemStatement):
    from vizier_amd.converters.core import TrialToArrayConverter
    self._model = model
    self._problem = problem
    self._converter = TrialToArrayConverter(problem)
    self._metric = problem.metric_information.item().name

  def evaluate(self, suggestions: Sequence[vz.Trial]) -> None:
    feats = self._converter.to_features(suggestions)
    values = np.asarray(self._model.predict(feats)).reshape(-1)
    for trial, v in zip(suggestions, values):
      trial.complete(vz.Measurement(metrics={self._metric: float(v)}))

  def problem_statement(self) -> vz.ProblemStatement:
    return copy.deepcopy(self._problem)
