"""Classifier wrappers for algorithm research.

Capability parity with vizier/_src/algorithms/classification/
classifiers.py (SklearnClassifier :32): a thin, typed wrapper around any
sklearn-style classifier for good/bad trial discrimination.
"""

from __future__ import annotations

from typing import Optional, Sequence

import numpy as np


class SklearnClassifier:
  """Wraps an sklearn classifier with feature/label plumbing."""

  def __init__(self, classifier, *, features_train: np.ndarray,
               labels_train: np.ndarray, features_test: np.ndarray,
               eval_metric: str = 'probability'):
    self._classifier = classifier
    self._x_train = np.asarray(features_train)
    self._y_train = np.asarray(labels_train)
    self._x_test = np.asarray(features_test)
    if eval_metric not in ('probability', 'decision'):
      raise ValueError(f'Unknown eval_metric {eval_metric}')
    self._eval_metric = eval_metric

  def __call__(self) -> np.ndarray:
    self._classifier.fit(self._x_train, self._y_train)
    if self._eval_metric == 'probability':
      probs = self._classifier.predict_proba(self._x_test)
      return probs[:, -1]
    return self._classifier.decision_function(self._x_test)
