"""Core tuning objects: VizierTrial, Feedback, Result.

Parity with vizier/_src/pyglove/core.py (VizierTrial :41, Feedback
:147, Result): DNA/metadata lazily materialize from the Vizier trial,
Feedback drives the pg.sample loop (add_measurement / done / skip /
should_stop_early) against a clients.Trial.
"""

from __future__ import annotations

from typing import Any, Dict, List, Optional, Sequence

import pyglove as pg

from vizier_amd import pyvizier as vz
from vizier_amd._src.pyglove import constants
from vizier_amd._src.pyglove import converters


def _reward_for_feedback(converter: converters.VizierConverter,
                         trial: vz.Trial,
                         metric_names: Sequence[str]) -> Optional[float]:
  """Final-measurement reward tuple head (core.py get_reward_for...)."""
  if trial.final_measurement is None or trial.infeasible:
    return None
  metrics = trial.final_measurement.metrics
  values = []
  for name in metric_names:
    raw = name[len('negative_'):] if name.startswith('negative_') else \
        name
    m = metrics.get(raw)
    if m is None:
      return None
    values.append(-m.value if name.startswith('negative_') else m.value)
  return float(values[0]) if values else None


class VizierTrial:
  """pg.tuning.Trial-shaped view over a vz.Trial (core.py:41)."""

  def __init__(self, converter: converters.VizierConverter,
               trial: vz.Trial):
    self._converter = converter
    self._trial = trial
    self._dna = None

  @property
  def id(self) -> int:
    return self._trial.id

  @property
  def status(self) -> str:
    # PENDING is the legacy name for ACTIVE (core.py:34).
    status = self._trial.status
    return 'PENDING' if status == vz.TrialStatus.ACTIVE else status.value

  @property
  def infeasible(self) -> bool:
    return self._trial.infeasible

  @property
  def dna(self):
    if self._dna is None:
      self._dna = self._converter.to_dna(self._trial)
    return self._dna

  @property
  def metadata(self) -> Dict[str, Any]:
    return converters.get_pyglove_metadata(self._trial)

  @property
  def measurements(self) -> List[vz.Measurement]:
    return list(self._trial.measurements)

  def get_reward_for_feedback(self, metric_names: Sequence[str]
                              ) -> Optional[float]:
    return _reward_for_feedback(self._converter, self._trial,
                                metric_names)


class Feedback:
  """Tuning feedback over a clients.Trial (core.py:147)."""

  def __init__(self, vizier_trial, converter: converters.VizierConverter):
    self._trial_client = vizier_trial
    self._converter = converter
    self._trial = vizier_trial.materialize()

  @property
  def id(self) -> int:
    return self._trial_client.id

  @property
  def dna(self):
    return self._converter.to_dna(self._trial)

  def get_trial(self) -> VizierTrial:
    self._trial = self._trial_client.materialize()
    return VizierTrial(self._converter, self._trial)

  def add_measurement(self, reward: Optional[float] = None, *,
                      metrics: Optional[Dict[str, float]] = None,
                      step: Optional[int] = None,
                      elapse_secs: Optional[float] = None,
                      done: bool = False) -> None:
    all_metrics = dict(metrics or {})
    if reward is not None:
      for name in self._converter.metrics_to_optimize:
        raw = name[len('negative_'):] if name.startswith('negative_') \
            else name
        all_metrics.setdefault(
            raw, -reward if name.startswith('negative_') else reward)
    measurement = vz.Measurement(metrics=all_metrics)
    if step is not None:
      measurement.steps = step
    if elapse_secs is not None:
      measurement.elapsed_secs = elapse_secs
    if done:
      self._trial_client.complete(measurement)
    else:
      self._trial_client.add_measurement(measurement)

  def done(self, metadata: Optional[Dict[str, Any]] = None) -> None:
    del metadata
    self._trial_client.complete()

  def skip(self, reason: Optional[str] = None) -> None:
    self._trial_client.complete(
        vz.Measurement(),
        infeasible_reason=reason or 'skipped by feedback')

  def should_stop_early(self) -> bool:
    return self._trial_client.check_early_stopping()

  def __call__(self, reward: Optional[float] = None, **metrics) -> None:
    self.add_measurement(reward, metrics=metrics or None, done=True)


class Result:
  """Study-level tuning result (core.py Result.from_study)."""

  def __init__(self, trials: List[VizierTrial],
               converter: converters.VizierConverter):
    self._trials = trials
    self._converter = converter

  @classmethod
  def from_study(cls, study) -> 'Result':
    problem = study.materialize_problem_statement()
    converter = converters.VizierConverter.from_problem(problem)
    trials = [VizierTrial(converter, t)
              for t in study.trials().get()]
    return cls(trials, converter)

  @property
  def trials(self) -> List[VizierTrial]:
    return list(self._trials)

  def best_trial(self) -> Optional[VizierTrial]:
    metric_names = self._converter.metrics_to_optimize
    best, best_reward = None, None
    for t in self._trials:
      r = t.get_reward_for_feedback(metric_names)
      if r is not None and (best_reward is None or r > best_reward):
        best, best_reward = t, r
    return best
