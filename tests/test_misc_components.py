"""Tests: regression, classification, demos, integration shims."""

import subprocess
import sys

import numpy as np
import pytest

from vizier_amd import pyvizier as vz
from vizier_amd._src.algorithms.classification import SklearnClassifier
from vizier_amd._src.algorithms.regression import (
    GBMAutoRegressor,
    trials_to_curves,
)


class TestRegression:

  def test_curve_extraction(self):
    t = vz.Trial(id=1)
    t.measurements = [
        vz.Measurement(metrics={'loss': 3.0}, steps=1),
        vz.Measurement(metrics={'loss': 2.0}, steps=2),
        vz.Measurement(metrics={'loss': 1.5}, steps=3),
    ]
    curves = trials_to_curves([t], 'loss')
    np.testing.assert_allclose(curves[0], [3.0, 2.0, 1.5])

  def test_gbm_autoregressor_learns_decay(self):
    rng = np.random.default_rng(0)
    curves = []
    for _ in range(30):
      start = rng.uniform(1.0, 3.0)
      steps = np.arange(20)
      curves.append(start * 0.9 ** steps)
    model = GBMAutoRegressor(window=3, seed=0).fit(curves)
    prefix = 2.0 * 0.9 ** np.arange(8)
    pred = model.predict_final(prefix, total_steps=20)
    true = 2.0 * 0.9 ** 19
    assert abs(pred - true) < 0.1


class TestClassification:

  def test_sklearn_wrapper(self):
    from sklearn.linear_model import LogisticRegression
    rng = np.random.default_rng(1)
    x = rng.standard_normal((100, 2))
    y = (x[:, 0] > 0).astype(int)
    test = np.array([[3.0, 0.0], [-3.0, 0.0]])
    probs = SklearnClassifier(LogisticRegression(),
                              features_train=x, labels_train=y,
                              features_test=test)()
    assert probs[0] > 0.9 and probs[1] < 0.1


class TestDemos:

  def test_client_demo_runs_in_process(self):
    result = subprocess.run(
        [sys.executable, 'demos/run_vizier_client.py',
         '--max_num_iterations', '2', '--algorithm', 'RANDOM_SEARCH'],
        capture_output=True, text=True, timeout=120)
    assert result.returncode == 0, result.stderr
    assert 'Optimal trial' in result.stdout


class TestIntegrationShims:

  def test_raytune_module_importable_without_ray(self):
    from vizier_amd._src.raytune import vizier_search
    searcher = vizier_search.VizierSearch(algorithm='RANDOM_SEARCH')
    assert searcher is not None

  def test_pyglove_module_importable_without_pyglove(self):
    from vizier_amd._src.pyglove import vizier_backend
    with pytest.raises(ImportError):
      vizier_backend._require_pyglove()
