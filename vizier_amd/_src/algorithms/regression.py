"""Trial-curve regression for early-stopping research.

Capability parity with vizier/_src/algorithms/regression/
trial_regression_utils.py (GBMAutoRegressor :165): gradient-boosted
auto-regression of intermediate measurement curves, used to hallucinate
a trial's final value from its prefix.
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


class GBMAutoRegressor:
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
