"""HPO-B benchmark handler + experimenter (offline-data-driven).

Capability parity with
vizier/_src/benchmarks/experimenters/hpob/handler.py (HPOBHandler :36):
loads the HPO-B meta-dataset JSON files (meta-test / meta-train /
meta-validation / bo-initializations), normalizes targets, and exposes
the same tabular `evaluate` loop. The continuous mode in the reference
queries saved XGBoost surrogates; offline (no xgboost) this module
fits a scikit-learn gradient-boosted surrogate from the tabular (X, y)
of the chosen dataset instead — same interface, documented substitute.

Data is NOT bundled (the real HPO-B download is ~1 GB): everything
degrades gracefully — `HPOBHandler.is_available(root_dir)` says whether
the files exist, and construction raises FileNotFoundError with the
download pointer when they do not. Tests exercise the full code path
through a tiny in-repo fixture (tests/data/hpob).
"""

from __future__ import annotations

import functools
import json
import os
from typing import Dict, List, Optional, Sequence

import numpy as np

from vizier_amd import pyvizier as vz
from vizier_amd._src.benchmarks.experimenters import experimenter

_META_TEST = 'meta-test-dataset.json'
_META_TRAIN = 'meta-train-dataset.json'
_META_TRAIN_AUG = 'meta-train-dataset-augmented.json'
_META_VALID = 'meta-validation-dataset.json'
_BO_INIT = 'bo-initializations.json'


class HPOBHandler:
  """Loads and serves the HPO-B meta-dataset (handler.py:36)."""

  SEEDS = ['test0', 'test1', 'test2', 'test3', 'test4']

  def __init__(self, root_dir: str = 'hpob-data/',
               mode: str = 'v3-test'):
    if not self.is_available(root_dir):
      raise FileNotFoundError(
          f'HPO-B data not found under {root_dir!r}. Download the '
          'benchmark data (hpob-data: meta-*.json + '
          'bo-initializations.json) from the HPO-B release '
          '(github.com/releaunifreiburg/HPO-B) and point root_dir at '
          'it.')
    self.mode = mode
    self.seeds = list(self.SEEDS)
    if mode == 'v3-test':
      self._load(root_dir, only_test=True)
    elif mode == 'v3-train-augmented':
      self._load(root_dir, only_test=False, augmented=True)
    elif mode in ('v1', 'v2', 'v3'):
      self._load(root_dir, only_test=False, merge=(mode != 'v3'))
    else:
      raise ValueError(f'Invalid mode {mode!r}')

  @staticmethod
  def is_available(root_dir: str) -> bool:
    return (os.path.exists(os.path.join(root_dir, _META_TEST)) and
            os.path.exists(os.path.join(root_dir, _BO_INIT)))

  def _load(self, root: str, *, only_test: bool,
            augmented: bool = False, merge: bool = False) -> None:
    with open(os.path.join(root, _META_TEST), 'rb') as f:
      self.meta_test_data = json.load(f)
    with open(os.path.join(root, _BO_INIT), 'rb') as f:
      self.bo_initializations = json.load(f)
    self.meta_train_data: Dict = {}
    self.meta_validation_data: Dict = {}
    if not only_test:
      train_name = _META_TRAIN_AUG if augmented else _META_TRAIN
      train_path = os.path.join(root, train_name)
      valid_path = os.path.join(root, _META_VALID)
      if os.path.exists(train_path):
        with open(train_path, 'rb') as f:
          self.meta_train_data = json.load(f)
      if os.path.exists(valid_path):
        with open(valid_path, 'rb') as f:
          self.meta_validation_data = json.load(f)
      if merge:
        merged: Dict = {}
        for split in (self.meta_train_data, self.meta_test_data,
                      self.meta_validation_data):
          for space, datasets in split.items():
            merged.setdefault(space, {}).update(datasets)
        self.meta_test_data = merged

  @staticmethod
  def normalize(y: np.ndarray, y_min: Optional[float] = None,
                y_max: Optional[float] = None) -> np.ndarray:
    if y_min is None:
      y_min, y_max = float(np.min(y)), float(np.max(y))
    span = max(y_max - y_min, 1e-12)
    return (y - y_min) / span

  def dataset(self, search_space_id: str, dataset_id: str):
    entry = self.meta_test_data[search_space_id][dataset_id]
    return np.asarray(entry['X'], dtype=np.float64), \
        np.asarray(entry['y'], dtype=np.float64).reshape(-1)

  def get_seeds(self) -> List[str]:
    return list(self.seeds)

  def evaluate(self, bo_method=None, search_space_id: str = None,
               dataset_id: str = None, seed: str = None,
               n_trials: int = 10) -> List[float]:
    """Tabular evaluation loop (handler.py:170-231): the method picks
    among the pending rows; returns the incumbent history."""
    assert bo_method is not None and hasattr(bo_method,
                                             'observe_and_suggest')
    X, y = self.dataset(search_space_id, dataset_id)
    y = self.normalize(y)
    pending = list(range(len(X)))
    current: List[int] = []
    init_ids = self.bo_initializations[search_space_id][dataset_id][seed]
    for i in range(5):
      pending.remove(init_ids[i])
      current.append(init_ids[i])
    history = [float(np.max(y[current]))]
    for _ in range(n_trials):
      pick = bo_method.observe_and_suggest(X[current], y[current],
                                           X[pending])
      idx = pending[pick]
      pending.remove(idx)
      current.append(idx)
      history.append(float(np.max(y[current])))
    return history


@functools.lru_cache(maxsize=32)
def _fit_surrogate(root_dir: str, mode: str, search_space_id: str,
                   dataset_id: str, seed: int):
  from sklearn import ensemble
  handler = HPOBHandler(root_dir, mode)
  X, y = handler.dataset(search_space_id, dataset_id)
  model = ensemble.GradientBoostingRegressor(random_state=seed)
  model.fit(X, handler.normalize(y))
  return model, X.shape[1]


class HPOBExperimenter(experimenter.Experimenter):
  """Continuous-space experimenter over one HPO-B (space, dataset).

  The search space is the unit hypercube of the tabular feature matrix
  (HPO-B stores all configurations pre-scaled to [0,1]); evaluation
  scores a suggestion through a surrogate regressor fit on the tabular
  rows (sklearn GBM offline; the reference loads saved XGBoost
  surrogates, handler.py:253-260).
  """

  def __init__(self, root_dir: str, search_space_id: str,
               dataset_id: str, *, mode: str = 'v3-test', seed: int = 0):
    self._ids = (search_space_id, dataset_id)
    self._model, self._dim = _fit_surrogate(
        root_dir, mode, search_space_id, dataset_id, seed)

  def problem_statement(self) -> vz.ProblemStatement:
    problem = vz.ProblemStatement()
    for i in range(self._dim):
      problem.search_space.root.add_float_param(f'x{i}', 0.0, 1.0)
    problem.metric_information.append(vz.MetricInformation(
        name='accuracy', goal=vz.ObjectiveMetricGoal.MAXIMIZE))
    return problem

  def evaluate(self, suggestions: Sequence[vz.Trial]) -> None:
    for trial in suggestions:
      x = np.array([[trial.parameters.get_value(f'x{i}')
                     for i in range(self._dim)]])
      value = float(self._model.predict(x)[0])
      trial.complete(vz.Measurement(metrics={'accuracy': value}))

  def __repr__(self) -> str:
    return f'HPOBExperimenter{self._ids}'
