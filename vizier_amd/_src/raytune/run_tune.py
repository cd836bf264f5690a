"""Ray Tune entry points over Vizier experimenters.

Capability parity with vizier/_src/raytune/run_tune.py
(run_tune_distributed :32, run_tune_bbob :53, run_tune_from_factory
:85) and converters.py (SearchSpaceConverter.to_dict :31,
ExperimenterConverter.to_callable :113). Ray is an optional
dependency: imports are deferred so the module (and callables that
don't need Ray) work without it.
"""

from __future__ import annotations

from typing import Any, Callable, Dict, List, Optional, Tuple

import numpy as np

from vizier_amd import pyvizier as vz
from vizier_amd._src.benchmarks.experimenters.experimenter import (
    Experimenter,
)


class SearchSpaceConverter:
  """pyvizier.SearchSpace <-> Ray Tune param_space dict."""

  @classmethod
  def to_dict(cls, search_space: vz.SearchSpace) -> Dict[str, Any]:
    from ray import tune  # Deferred: ray is optional.
    param_space: Dict[str, Any] = {}
    for param in search_space.parameters:
      if param.type == vz.ParameterType.DOUBLE:
        lo, hi = param.bounds
        if param.scale_type == vz.ScaleType.LOG:
          param_space[param.name] = tune.loguniform(lo, hi)
        else:
          param_space[param.name] = tune.uniform(lo, hi)
      elif param.type == vz.ParameterType.INTEGER:
        lo, hi = param.bounds
        param_space[param.name] = tune.randint(int(lo), int(hi) + 1)
      elif param.type in (vz.ParameterType.CATEGORICAL,
                          vz.ParameterType.DISCRETE):
        param_space[param.name] = tune.choice(
            list(param.feasible_values))
      else:
        raise ValueError(f'Unsupported parameter {param}')
    return param_space

  # Inverse direction lives in vizier_search.SearchSpaceConverter
  # (to_vizier); re-exported here for a single entry point.
  @classmethod
  def to_vizier(cls, param_space: Dict[str, Any]) -> vz.SearchSpace:
    from vizier_amd._src.raytune.vizier_search import (
        SearchSpaceConverter as _Inverse,
    )
    return _Inverse.to_vizier(param_space)


class ExperimenterConverter:
  """Experimenter -> Ray Tune trainable callable (ray-free)."""

  @classmethod
  def to_callable(cls, experimenter: Experimenter
                  ) -> Callable[[Dict[str, Any]], Dict[str, float]]:
    def trainable(config: Dict[str, Any]) -> Dict[str, float]:
      trial = vz.Trial(parameters=dict(config))
      experimenter.evaluate([trial])
      return {name: metric.value
              for name, metric in trial.final_measurement.metrics.items()}
    return trainable


def run_tune_from_factory(experimenter_factory, tune_config=None,
                          run_config=None):
  """Runs a Ray Tuner over an ExperimenterFactory's problem."""
  from ray import tune
  from ray.air import session

  experimenter = experimenter_factory()
  problem = experimenter.problem_statement()
  param_space = SearchSpaceConverter.to_dict(problem.search_space)
  objective = ExperimenterConverter.to_callable(experimenter)
  metric_info = problem.metric_information.item()
  if tune_config is None:
    tune_config = tune.TuneConfig()
  tune_config.metric = metric_info.name
  tune_config.mode = ('min' if metric_info.goal ==
                      vz.ObjectiveMetricGoal.MINIMIZE else 'max')

  def objective_fn(config) -> None:
    for _ in range(tune_config.num_samples):
      session.report(objective(config))

  tuner = tune.Tuner(objective_fn, param_space=param_space,
                     run_config=run_config, tune_config=tune_config)
  return tuner.fit()


def run_tune_bbob(function_name: str, dimension: int,
                  shift: Optional[np.ndarray] = None, tune_config=None,
                  run_config=None):
  """Runs a Ray Tuner on a (shifted) BBOB function."""
  from vizier_amd._src.benchmarks.experimenters.experimenter_factory import (
      BBOBExperimenterFactory,
  )
  from vizier_amd._src.benchmarks.experimenters.wrappers import (
      ShiftingExperimenter,
  )
  factory = BBOBExperimenterFactory(name=function_name, dim=dimension)
  if shift is not None:
    base_factory = factory
    factory = lambda: ShiftingExperimenter(base_factory(), shift=shift)
  return run_tune_from_factory(factory, tune_config, run_config)


def run_tune_distributed(run_tune_args_list: List[Tuple[Any, ...]],
                         run_tune: Callable[..., Any]) -> List[Any]:
  """Fans run_tune calls out via Ray's datasets API (map over args)."""
  import ray

  @ray.remote
  def _one(args):
    return run_tune(*args)

  return ray.get([_one.remote(args) for args in run_tune_args_list])
