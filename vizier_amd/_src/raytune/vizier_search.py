"""Ray Tune integration: a Searcher backed by this Vizier service.

Capability parity with vizier/_src/raytune/vizier_search.py:31
(VizierSearch) and converters.py (SearchSpaceConverter). Requires the
`ray` package at call time (not installed in this image; imports are
deferred so the module itself is always importable).
"""

from __future__ import annotations

from typing import Any, Dict, Optional

from vizier_amd import pyvizier as vz
from vizier_amd.service import clients


class SearchSpaceConverter:
  """Ray Tune param_space dict -> vz.SearchSpace."""

  @classmethod
  def to_vizier(cls, param_space: Dict[str, Any]) -> vz.SearchSpace:
    from ray.tune.search import sample  # Deferred: ray is optional.
    space = vz.SearchSpace()
    root = space.root
    for name, dist in param_space.items():
      if isinstance(dist, sample.Float):
        is_log = isinstance(getattr(dist, 'sampler', None),
                            sample.LogUniform)
        root.add_float_param(
            name, dist.lower, dist.upper,
            scale_type=vz.ScaleType.LOG if is_log
            else vz.ScaleType.LINEAR)
      elif isinstance(dist, sample.Integer):
        root.add_int_param(name, dist.lower, dist.upper - 1)
      elif isinstance(dist, sample.Categorical):
        values = dist.categories
        if all(isinstance(v, (int, float)) and not isinstance(v, bool)
               for v in values):
          root.add_discrete_param(name, values)
        else:
          root.add_categorical_param(name, [str(v) for v in values])
      else:
        raise ValueError(f'Unsupported Ray Tune distribution for '
                         f'{name}: {dist!r}')
    return space


class VizierSearch:
  """ray.tune.search.Searcher implementation over clients.Study.

  Instantiate and pass to `tune.Tuner(..., tune_config=TuneConfig(
  search_alg=VizierSearch(...)))`. Implemented against the public
  Searcher protocol (set_search_properties / suggest / on_trial_complete).
  """

  def __init__(self, study_id: Optional[str] = None,
               algorithm: str = 'DEFAULT', *,
               owner: str = 'raytune',
               metric: Optional[str] = None, mode: Optional[str] = None):
    self._study_id = study_id or 'raytune_study'
    self._owner = owner
    self._algorithm = algorithm
    self._metric = metric
    self._mode = mode
    self._study: Optional[clients.Study] = None
    self._ray_trial_to_vizier: Dict[str, int] = {}

  def set_search_properties(self, metric: Optional[str],
                            mode: Optional[str],
                            config: Dict[str, Any], **spec) -> bool:
    self._metric = metric or self._metric or 'objective'
    self._mode = mode or self._mode or 'max'
    goal = (vz.ObjectiveMetricGoal.MAXIMIZE if self._mode == 'max'
            else vz.ObjectiveMetricGoal.MINIMIZE)
    study_config = vz.StudyConfig(
        search_space=SearchSpaceConverter.to_vizier(config),
        metric_information=[vz.MetricInformation(name=self._metric,
                                                 goal=goal)],
        algorithm=self._algorithm)
    self._study = clients.Study.from_study_config(
        study_config, owner=self._owner, study_id=self._study_id)
    return True

  def suggest(self, trial_id: str) -> Optional[Dict[str, Any]]:
    if self._study is None:
      raise RuntimeError('set_search_properties must be called first.')
    suggestions = self._study.suggest(count=1, client_id=trial_id)
    if not suggestions:
      return None
    trial = suggestions[0]
    self._ray_trial_to_vizier[trial_id] = trial.id
    return dict(trial.parameters)

  def on_trial_complete(self, trial_id: str,
                        result: Optional[Dict[str, Any]] = None,
                        error: bool = False) -> None:
    vizier_id = self._ray_trial_to_vizier.pop(trial_id, None)
    if vizier_id is None or self._study is None:
      return
    trial_client = self._study.get_trial(vizier_id)
    if error or result is None or self._metric not in result:
      trial_client.complete(infeasible_reason='ray trial error')
    else:
      trial_client.complete(vz.Measurement(
          metrics={self._metric: float(result[self._metric])}))

  def on_trial_result(self, trial_id: str,
                      result: Dict[str, Any]) -> None:
    vizier_id = self._ray_trial_to_vizier.get(trial_id)
    if vizier_id is None or self._study is None:
      return
    if self._metric in result:
      self._study.get_trial(vizier_id).add_measurement(vz.Measurement(
          metrics={self._metric: float(result[self._metric])},
          steps=int(result.get('training_iteration', 0))))
