"""Trial-curve regression for early-stopping research.

Capability parity with vizier/_src/algorithms/regression/
trial_regression_utils.py: TrialData curve extraction with step
extrapolation / dedupe / linear interpolation (:41-163), the
target-step GBM auto-regressor with lag features and CV grid search
(GBMAutoRegressor :165), and the GBMTrialHallucinator (:366) that
completes stopped trials with hallucinated final measurements. The GBM
backend is sklearn's GradientBoostingRegressor (the reference uses
lightGBM, unavailable offline — same algorithm family).

`WindowedAutoRegressor` is an additional roll-forward variant (window
of k past values -> next value), kept for curve research.
"""

from __future__ import annotations

from typing import List, Optional, Sequence, Tuple

import numpy as np
from sklearn import ensemble

from vizier_amd import pyvizier as vz


def trials_to_curves(trials: Sequence[vz.Trial], metric: str
                     ) -> List[np.ndarray]:
  """Extracts per-trial intermediate metric curves (in step order)."""
  curves = []
  for t in trials:
    points = [(m.steps, m.metrics[metric].value)
              for m in t.measurements if metric in m.metrics]
    points.sort()
    curves.append(np.asarray([v for _, v in points], dtype=np.float64))
  return curves


class WindowedAutoRegressor:
  """Predicts curve continuations with gradient-boosted trees.

  Trains on (window of k past values -> next value) pairs pooled over
  all training curves; prediction rolls the model forward.
  """

  def __init__(self, window: int = 5, *, n_estimators: int = 100,
               learning_rate: float = 0.1, max_depth: int = 3,
               seed: Optional[int] = None):
    self.window = window
    self._model = ensemble.GradientBoostingRegressor(
        n_estimators=n_estimators, learning_rate=learning_rate,
        max_depth=max_depth, random_state=seed)
    self._fitted = False

  def fit(self, curves: Sequence[np.ndarray]) -> 'GBMAutoRegressor':
    xs, ys = [], []
    for curve in curves:
      for i in range(len(curve) - self.window):
        xs.append(curve[i:i + self.window])
        ys.append(curve[i + self.window])
    if not xs:
      raise ValueError(
          f'Need curves longer than window={self.window} to fit.')
    self._model.fit(np.stack(xs), np.asarray(ys))
    self._fitted = True
    return self

  def predict_next(self, prefix: np.ndarray) -> float:
    if not self._fitted:
      raise ValueError('fit() first.')
    if len(prefix) < self.window:
      raise ValueError(f'Prefix must have >= {self.window} points.')
    return float(self._model.predict(
        prefix[-self.window:][None, :])[0])

  def predict_final(self, prefix: np.ndarray, total_steps: int
                    ) -> float:
    """Rolls the auto-regression forward to `total_steps` points."""
    curve = list(np.asarray(prefix, dtype=np.float64))
    while len(curve) < total_steps:
      curve.append(self.predict_next(np.asarray(curve)))
    return float(curve[total_steps - 1])


# -- Reference-parity target-step machinery ---------------------------------


import dataclasses
from typing import Any, Dict

from scipy.interpolate import InterpolatedUnivariateSpline
from sklearn import model_selection


@dataclasses.dataclass
class TrialData:
  """Lightweight curve record (trial_regression_utils.py:41)."""

  id: int
  learning_rate: float
  final_objective: float
  steps: List[float]
  objective_values: List[float]

  @classmethod
  def from_trial(cls, trial: vz.Trial, *, learning_rate_param_name: str,
                 metric_name: str, use_steps: bool = True,
                 goal_is_maximize: bool = True) -> 'TrialData':
    lr = trial.parameters.get_value(learning_rate_param_name, 0.0)
    points = []
    for m in trial.measurements:
      if metric_name in m.metrics:
        t = m.steps if use_steps else m.elapsed_secs
        points.append((float(t or 0), float(m.metrics[metric_name].value)))
    points.sort(key=lambda p: p[0])
    steps = [p[0] for p in points]
    values = [p[1] for p in points]
    if (trial.final_measurement is not None and
        metric_name in trial.final_measurement.metrics):
      final = float(trial.final_measurement.metrics[metric_name].value)
    else:
      final = values[-1] if values else 0.0
    return cls(id=trial.id or 0, learning_rate=float(lr),
               final_objective=final, steps=steps,
               objective_values=values)

  def extrapolate_trial_objective_value(self, max_num_steps: float) -> None:
    """Holds the last value out to `max_num_steps` (:97)."""
    if not self.steps or self.steps[-1] >= max_num_steps:
      return
    self.steps.append(max_num_steps)
    self.objective_values.append(self.objective_values[-1])


def sort_dedupe_measurements(steps: List[float], values: List[float]
                             ) -> Tuple[List[float], List[float]]:
  """Keeps the LAST value per step, strictly increasing steps (:134)."""
  by_step: Dict[float, float] = {}
  for s, v in zip(steps, values):
    by_step[s] = v
  out_s, out_v = [], []
  for s in sorted(by_step):
    out_s.append(s)
    out_v.append(by_step[s])
  return out_s, out_v


def interpolation_fn(steps: List[float], values: List[float]):
  """Linear spline through the curve (:112)."""
  if len(steps) == 1:
    return lambda t: values[0]
  return InterpolatedUnivariateSpline(steps, values, k=1)


class GBMAutoRegressor:
  """Target-step GBM auto-regressor (trial_regression_utils.py:165).

  Features per training row: [learning_rate] + min_points x
  (target_step - step_j, value_j) lags; the target is the curve's
  interpolated value AT target_step. Model selection via k-fold
  GridSearchCV over a small depth/estimators grid.
  """

  def __init__(self, target_step: float, min_points: int, *,
               learning_rate_param_name: str = 'learning_rate',
               metric_name: str = 'objective',
               use_steps: bool = True,
               gbdt_param_grid: Optional[Dict[str, Any]] = None,
               cv: int = 2, random_state: Optional[int] = None):
    self._target_step = target_step
    self._min_points = min_points
    self._lr_name = learning_rate_param_name
    self._metric_name = metric_name
    self._use_steps = use_steps
    self._grid = gbdt_param_grid or {'max_depth': [2, 3, 5],
                                     'n_estimators': [50, 100]}
    self._cv = cv
    self._random_state = random_state
    self._model = None
    self.best_params: Optional[Dict[str, Any]] = None

  @property
  def is_trained(self) -> bool:
    return self._model is not None

  def _features(self, td: TrialData, end_index: int) -> List[float]:
    if self._min_points > end_index + 1:
      raise ValueError('Not enough data before end_index.')
    feats = [td.learning_rate]
    for j in range(self._min_points):
      feats.append(self._target_step - td.steps[end_index - j])
      feats.append(td.objective_values[end_index - j])
    return feats

  def train(self, trials: Sequence[vz.Trial]) -> None:
    data = [TrialData.from_trial(
        t, learning_rate_param_name=self._lr_name,
        metric_name=self._metric_name, use_steps=self._use_steps)
        for t in trials]
    if len(data) < self._min_points + 1:
      return
    xs, ys = [], []
    for td in data:
      if len(td.steps) < self._min_points + 1:
        continue
      td.extrapolate_trial_objective_value(self._target_step)
      s, v = sort_dedupe_measurements(td.steps, td.objective_values)
      interp = interpolation_fn(s, v)
      target_value = float(interp(self._target_step))
      for i, step in enumerate(td.steps):
        if i < self._min_points - 1 or step >= self._target_step:
          continue
        xs.append(self._features(td, i))
        ys.append(target_value)
    if not xs:
      return
    x = np.asarray(xs)
    if x.shape[0] <= (self._min_points + 1) / (1.0 - 1.0 / self._cv):
      return
    cv = model_selection.GridSearchCV(
        ensemble.GradientBoostingRegressor(
            random_state=self._random_state),
        self._grid, cv=self._cv)
    cv.fit(x, np.asarray(ys))
    self.best_params = dict(cv.best_params_)
    self._model = ensemble.GradientBoostingRegressor(
        random_state=self._random_state, **self.best_params
    ).fit(x, np.asarray(ys))

  def predict(self, trial: vz.Trial) -> Optional[float]:
    if not self.is_trained:
      raise ValueError('Prediction requires a trained model.')
    td = TrialData.from_trial(
        trial, learning_rate_param_name=self._lr_name,
        metric_name=self._metric_name, use_steps=self._use_steps)
    if len(td.steps) < self._min_points:
      return None
    feats = np.asarray(
        self._features(td, len(td.steps) - 1)).reshape(1, -1)
    return float(self._model.predict(feats)[0])


@dataclasses.dataclass
class HallucinationOptions:
  """Reference trial_regression_utils.py:333."""

  autoregressive_order: int = 5
  learning_rate_param_name: str = 'learning_rate'
  use_steps: bool = True
  gbdt_param_grid: Optional[Dict[str, Any]] = None
  min_completed_trials: int = 5
  min_steps: int = 5
  max_steps: Optional[int] = None
  random_state: Optional[int] = None
  elapsed_seconds_gap: float = 0.0


class GBMTrialHallucinator:
  """Completes stopped trials with GBM-predicted final measurements
  (trial_regression_utils.py:366)."""

  def __init__(self, problem: vz.ProblemStatement,
               options: Optional[HallucinationOptions] = None):
    metrics = list(problem.metric_information)
    if len(metrics) != 1:
      raise ValueError('Single-objective only.')
    self._metric = metrics[0]
    self._options = options or HallucinationOptions()
    self._max_steps = self._options.max_steps
    self._model: Optional[GBMAutoRegressor] = None

  @property
  def is_trained(self) -> bool:
    return self._model is not None

  def train(self, trials: Sequence[vz.Trial]) -> None:
    opts = self._options
    if len(trials) < max(opts.autoregressive_order + 1,
                         opts.min_completed_trials):
      return
    if self._max_steps is None:
      self._max_steps = int(np.percentile(
          [len(t.measurements) for t in trials], 95))
    if not self._max_steps:
      return
    model = GBMAutoRegressor(
        target_step=self._max_steps,
        min_points=opts.autoregressive_order,
        learning_rate_param_name=opts.learning_rate_param_name,
        metric_name=self._metric.name, use_steps=opts.use_steps,
        gbdt_param_grid=opts.gbdt_param_grid,
        random_state=opts.random_state)
    model.train(trials)
    if model.is_trained:
      self._model = model

  def update_stopped_trials(self, stopped: List[vz.Trial]
                            ) -> List[vz.Trial]:
    if self._model is None:
      return stopped
    for trial in stopped:
      if trial.infeasible or trial.final_measurement is not None or \
          not trial.measurements:
        continue
      pred = self._model.predict(trial)
      if pred is None:
        continue
      import copy as _copy
      final = _copy.deepcopy(trial.measurements[-1])
      final.metrics[self._metric.name] = vz.Metric(value=pred)
      if self._options.use_steps:
        final.steps = self._max_steps
        final.elapsed_secs = (trial.measurements[-1].elapsed_secs or 0) + \
            self._options.elapsed_seconds_gap
      trial.complete(final)
    return stopped
