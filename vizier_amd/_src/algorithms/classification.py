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
    if classifier is None:
      raise ValueError('classifier must not be None')
    self._classifier = classifier
    self._x_train = np.asarray(features_train)
    self._y_train = np.asarray(labels_train).reshape(-1)
    self._x_test = np.asarray(features_test)
    if eval_metric not in ('probability', 'decision'):
      raise ValueError(f'Unknown eval_metric {eval_metric}')
    self._eval_metric = eval_metric

  def _validate(self) -> None:
    """Reference classifiers.py:54-83 input checks."""
    if self._x_train.ndim != 2 or self._x_test.ndim != 2:
      raise ValueError('features must be 2-D arrays')
    if self._x_train.shape[1] != self._x_test.shape[1]:
      raise ValueError('train/test feature dims differ: '
                       f'{self._x_train.shape} vs {self._x_test.shape}')
    if self._y_train.shape[0] != self._x_train.shape[0]:
      raise ValueError('labels/features row counts differ')
    values = np.unique(self._y_train)
    if not np.all(np.isin(values, (0, 1))):
      raise ValueError(f'labels must be binary 0/1, got {values}')

  def __call__(self) -> np.ndarray:
    self._validate()
    self._classifier.fit(self._x_train, self._y_train)
    if self._eval_metric == 'probability':
      probs = self._classifier.predict_proba(self._x_test)
      return probs[:, -1]
    return self._classifier.decision_function(self._x_test)
