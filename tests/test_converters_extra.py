"""feature_mapper / embedder / spatio_temporal converter tests."""

import numpy as np
import pytest

from vizier_amd import pyvizier as vz
from vizier_amd.converters.core import TrialToArrayConverter
from vizier_amd.converters import embedder, feature_mapper, spatio_temporal


def mixed_problem() -> vz.ProblemStatement:
  p = vz.ProblemStatement()
  root = p.search_space.root
  root.add_float_param('lr', 1e-4, 1e-1, scale_type=vz.ScaleType.LOG)
  root.add_categorical_param('opt', ['adam', 'sgd', 'lamb'])
  root.add_float_param('mom', 0.0, 1.0)
  root.add_int_param('layers', 1, 8)
  root.add_categorical_param('act', ['relu', 'gelu'])
  p.metric_information.append(vz.MetricInformation(
      name='acc', goal=vz.ObjectiveMetricGoal.MAXIMIZE))
  return p


class TestFeatureMapper:

  def test_map_unmap_roundtrip(self):
    problem = mixed_problem()
    conv = TrialToArrayConverter(problem)
    mapper = feature_mapper.ContinuousCategoricalFeatureMapper(conv)
    trials = [
        vz.TrialSuggestion({'lr': 1e-3, 'opt': 'sgd', 'mom': 0.9,
                            'layers': 4, 'act': 'gelu'}),
        vz.TrialSuggestion({'lr': 1e-2, 'opt': 'lamb', 'mom': 0.1,
                            'layers': 1, 'act': 'relu'}),
    ]
    feats = conv.to_features(trials)
    split = mapper.map(feats)
    assert split.continuous.shape == (2, 3)   # lr, mom, layers
    assert split.categorical.shape == (2, 2)  # opt, act
    # Categorical feasible values are SORTED by the config factory
    # (reference parameter_config semantics): opt -> [adam, lamb, sgd],
    # act -> [gelu, relu].
    assert split.categorical.tolist() == [[2, 0], [1, 1]]
    back = mapper.unmap(split)
    np.testing.assert_allclose(back, feats, atol=1e-12)

  def test_no_categoricals(self):
    p = vz.ProblemStatement()
    p.search_space.root.add_float_param('x', 0, 1)
    p.metric_information.append(vz.MetricInformation(
        name='m', goal=vz.ObjectiveMetricGoal.MAXIMIZE))
    conv = TrialToArrayConverter(p)
    mapper = feature_mapper.ContinuousCategoricalFeatureMapper(conv)
    split = mapper.map(conv.to_features([vz.TrialSuggestion({'x': .5})]))
    assert split.categorical.shape == (1, 0)
    assert mapper.n_categorical_params == 0


class TestEmbedder:

  def test_embedded_problem_statement(self):
    scaler = embedder.ProblemAndTrialsScaler(mixed_problem())
    space = scaler.problem_statement.search_space
    lr = space.get('lr')
    assert lr.type == vz.ParameterType.DOUBLE
    assert lr.bounds == (0.0, 1.0)
    layers = space.get('layers')
    assert layers.type == vz.ParameterType.DOUBLE
    assert layers.bounds == (0.0, 1.0)
    assert space.get('opt').type == vz.ParameterType.CATEGORICAL

  def test_map_unmap_roundtrip(self):
    problem = mixed_problem()
    scaler = embedder.ProblemAndTrialsScaler(problem)
    t = vz.Trial({'lr': 1e-2, 'opt': 'adam', 'mom': 0.25, 'layers': 3,
                  'act': 'relu'}, id=7)
    (m,) = scaler.map([t])
    assert isinstance(m, vz.Trial) and m.id == 7
    # log-scaled lr: 1e-2 in [1e-4, 1e-1] -> 2/3 of the log range.
    assert m.parameters.get_value('lr') == pytest.approx(2 / 3)
    assert m.parameters.get_value('opt') == 'adam'
    (u,) = scaler.unmap([m])
    assert u.parameters.get_value('lr') == pytest.approx(1e-2)
    assert u.parameters.get_value('layers') == 3
    assert u.parameters.get_value('act') == 'relu'

  def test_unmap_rounds_integers(self):
    problem = vz.ProblemStatement()
    problem.search_space.root.add_int_param('n', 0, 10)
    problem.metric_information.append(vz.MetricInformation(
        name='m', goal=vz.ObjectiveMetricGoal.MAXIMIZE))
    scaler = embedder.ProblemAndTrialsScaler(problem)
    (u,) = scaler.unmap([vz.TrialSuggestion({'n': 0.53})])
    assert u.parameters.get_value('n') == 5


def _timed_trial(values, steps=None):
  t = vz.Trial({'x': 0.5}, id=1)
  for i, v in enumerate(values):
    t.measurements.append(vz.Measurement(
        metrics={'m': v}, steps=steps[i] if steps else i + 1))
  return t


def _st_problem():
  p = vz.ProblemStatement()
  p.search_space.root.add_float_param('x', 0, 1)
  p.metric_information.append(vz.MetricInformation(
      name='m', goal=vz.ObjectiveMetricGoal.MAXIMIZE))
  return p


class TestTimedLabelsExtractor:

  SERIES = [2, 1, 0, 3, 3, 2, 4, 2, 1]

  def _extract(self, mode):
    ex = spatio_temporal.TimedLabelsExtractor(
        _st_problem().metric_information, 'index', value_extraction=mode)
    (tl,) = ex.convert([_timed_trial(self.SERIES)])
    return tl

  def test_raw(self):
    tl = self._extract(spatio_temporal.RAW)
    assert tl.labels['m'].reshape(-1).tolist() == self.SERIES

  def test_cummax(self):
    tl = self._extract(spatio_temporal.CUMMAX)
    assert tl.labels['m'].reshape(-1).tolist() == \
        [2, 2, 2, 3, 3, 3, 4, 4, 4]

  def test_cummax_lastonly(self):
    # Reference doc example: (2,1,0,3,3,2,4,2,1)->(_,_,2,_,_,3,_,_,4).
    tl = self._extract(spatio_temporal.CUMMAX_LASTONLY)
    assert tl.labels['m'].reshape(-1).tolist() == [2, 3, 4]
    assert tl.times.reshape(-1).tolist() == [2, 5, 8]

  def test_cummax_firstonly(self):
    # Reference doc example: ->(2,_,_,3,_,_,4,_,4).
    tl = self._extract(spatio_temporal.CUMMAX_FIRSTONLY)
    assert tl.labels['m'].reshape(-1).tolist() == [2, 3, 4, 4]
    assert tl.times.reshape(-1).tolist() == [0, 3, 6, 8]

  def test_minimize_goal_uses_cummin(self):
    p = _st_problem()
    p.metric_information = vz.MetricsConfig([vz.MetricInformation(
        name='m', goal=vz.ObjectiveMetricGoal.MINIMIZE)])
    ex = spatio_temporal.TimedLabelsExtractor(
        p.metric_information, 'index',
        value_extraction=spatio_temporal.CUMMAX)
    (tl,) = ex.convert([_timed_trial([3, 4, 1, 2])])
    assert tl.labels['m'].reshape(-1).tolist() == [3, 3, 1, 1]

  def test_steps_timestamps_and_all_timestamps(self):
    ex = spatio_temporal.TimedLabelsExtractor(
        _st_problem().metric_information, 'steps',
        value_extraction=spatio_temporal.RAW)
    t = _timed_trial([1, 2], steps=[10, 20])
    assert ex.extract_all_timestamps([t]) == [10.0, 20.0]


class TestSpatioTemporalConverters:

  def test_sparse_to_xy(self):
    problem = _st_problem()
    conv = TrialToArrayConverter(problem)
    ex = spatio_temporal.TimedLabelsExtractor(
        problem.metric_information, 'steps',
        value_extraction=spatio_temporal.RAW)
    sp = spatio_temporal.SparseSpatioTemporalConverter(conv, ex)
    t1 = _timed_trial([1.0, 2.0], steps=[1, 2])
    t2 = _timed_trial([5.0], steps=[3])
    x, y = sp.to_xy([t1, t2])
    assert x.shape == (3, 2) and y.shape == (3, 1)
    assert x[:, -1].tolist() == [1.0, 2.0, 3.0]  # timestamp column
    assert y.reshape(-1).tolist() == [1.0, 2.0, 5.0]
    feats = sp.to_features(vz.TrialSuggestion({'x': 0.5}), [1, 2, 3])
    assert feats.shape == (3, 2)

  def test_dense_to_xty(self):
    problem = _st_problem()
    conv = TrialToArrayConverter(problem)
    ex = spatio_temporal.TimedLabelsExtractor(
        problem.metric_information, 'steps',
        value_extraction=spatio_temporal.RAW)
    dn = spatio_temporal.DenseSpatioTemporalConverter(conv, ex)
    t1 = _timed_trial([1.0, 2.0], steps=[1, 2])
    t2 = _timed_trial([5.0], steps=[2])
    x, grid, y = dn.to_xty([t1, t2])
    assert x.shape == (2, 1)
    assert grid.tolist() == [1.0, 2.0]
    assert y[0].tolist() == [1.0, 2.0]
    assert np.isnan(y[1][0]) and y[1][1] == 5.0


class TestPaddingSchedule:
  """Parity with reference converters/padding.py:28-97 (3-axis buckets)."""

  def _trials(self, n):
    out = []
    for uid in range(1, n + 1):
      t = vz.Trial({'lr': 0.01, 'opt': 'adam', 'mom': 0.5,
                    'layers': 2, 'act': 'relu'}, id=uid)
      t.complete(vz.Measurement(metrics={'acc': float(uid)}))
      out.append(t)
    return out

  def test_three_axis_buckets(self):
    from vizier_amd.converters.core import PaddingSchedule, PaddingType
    s = PaddingSchedule(num_trials=PaddingType.MULTIPLES_OF_10,
                        num_features=PaddingType.POWERS_OF_2,
                        num_metrics=PaddingType.MULTIPLES_OF_10)
    conv = TrialToArrayConverter(mixed_problem(), padding_schedule=s)
    x, y, mask = conv.to_padded_xy(self._trials(13))
    assert x.shape[0] == 20 and y.shape[0] == 20
    assert mask.sum() == 13
    # 8 raw features (lr + 3 onehot + mom + layers + 2 onehot) -> 8
    # (already a power of 2); metrics 1 -> 10.
    assert x.shape[1] == 8
    assert y.shape[1] == 10
    assert np.isnan(y[:, 1:]).all()
    assert np.isnan(y[13:, 0]).all()

  def test_feature_padding_distance_neutral(self):
    # Zero-padded feature columns are identical across rows, so GP
    # pairwise distances are unchanged.
    import torch
    from vizier_amd.converters.core import PaddingSchedule, PaddingType
    from vizier_amd._src.gp.matern import gram_matern52
    s = PaddingSchedule(num_features=PaddingType.MULTIPLES_OF_10)
    conv_p = TrialToArrayConverter(mixed_problem(), padding_schedule=s)
    conv = TrialToArrayConverter(mixed_problem())
    trials = self._trials(5)
    xp, _, _ = conv_p.to_padded_xy(trials)
    x, _ = conv.to_xy(trials)
    assert xp.shape[1] == 10 and x.shape[1] == 8
    ls_p = torch.ones(10)
    ls = torch.ones(8)
    amp = torch.tensor(1.0)
    Kp = gram_matern52(torch.tensor(xp), None, ls_p, amp)
    K = gram_matern52(torch.tensor(x), None, ls, amp)
    assert torch.allclose(Kp, K, atol=1e-6)

  def test_bucket_stability_across_growth(self):
    from vizier_amd.converters.core import PaddingSchedule, PaddingType
    s = PaddingSchedule(num_trials=PaddingType.POWERS_OF_2)
    # 65..128 trials all land in the SAME padded shape: one hipGraph.
    sizes = {s.padded_size(n) for n in range(65, 129)}
    assert sizes == {128}
