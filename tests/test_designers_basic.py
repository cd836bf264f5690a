"""Tests for random / quasi-random / grid designers and policy wrappers."""

import numpy as np
import pytest

from vizier_amd import pyvizier as vz
from vizier_amd import pythia
from vizier_amd.algorithms import (
    DesignerPolicy,
    InRamPolicySupporter,
    PartiallySerializableDesignerPolicy,
)
from vizier_amd._src.algorithms.designers.grid import GridSearchDesigner
from vizier_amd._src.algorithms.designers.quasi_random import (
    QuasiRandomDesigner,
)
from vizier_amd._src.algorithms.designers.random import RandomDesigner


def flat_space() -> vz.SearchSpace:
  space = vz.SearchSpace()
  root = space.root
  root.add_float_param('x', -1.0, 2.0)
  root.add_float_param('lr', 1e-4, 1.0, scale_type=vz.ScaleType.LOG)
  root.add_int_param('i', 0, 9)
  root.add_categorical_param('c', ['r', 'g', 'b'])
  root.add_discrete_param('d', [0.5, 1.5])
  return space


def problem(space=None) -> vz.ProblemStatement:
  return vz.ProblemStatement(
      search_space=space or flat_space(),
      metric_information=[vz.MetricInformation(
          name='obj', goal=vz.ObjectiveMetricGoal.MAXIMIZE)])


def check_feasible(space: vz.SearchSpace, suggestions):
  for s in suggestions:
    for pc in space.parameters:
      assert pc.contains(s.parameters.get_value(pc.name)), \
          (pc.name, s.parameters.get_value(pc.name))


class TestRandomDesigner:

  def test_suggestions_feasible(self):
    space = flat_space()
    designer = RandomDesigner(space, seed=1)
    suggestions = designer.suggest(50)
    assert len(suggestions) == 50
    check_feasible(space, suggestions)

  def test_conditional_sampling(self):
    space = vz.SearchSpace()
    root = space.root
    root.add_categorical_param('model', ['dnn', 'linear'])
    root.select('model', ['dnn']).add_int_param('hidden', 1, 4)
    designer = RandomDesigner(space, seed=0)
    suggestions = designer.suggest(40)
    saw_child = saw_no_child = False
    for s in suggestions:
      if s.parameters.get_value('model') == 'dnn':
        assert 1 <= s.parameters.get_value('hidden') <= 4
        saw_child = True
      else:
        assert 'hidden' not in s.parameters
        saw_no_child = True
    assert saw_child and saw_no_child


class TestQuasiRandomDesigner:

  def test_low_discrepancy_better_than_iid_tail(self):
    space = vz.SearchSpace()
    space.root.add_float_param('x', 0.0, 1.0)
    designer = QuasiRandomDesigner(space, seed=5)
    xs = [s.parameters.get_value('x') for s in designer.suggest(128)]
    # Halton in 1-D: bin counts should be nearly uniform.
    counts, _ = np.histogram(xs, bins=8, range=(0, 1))
    assert counts.min() >= 12  # iid would frequently dip lower.

  def test_serialization_resumes_sequence(self):
    space = flat_space()
    d1 = QuasiRandomDesigner(space, seed=3)
    first = d1.suggest(5)
    state = d1.dump()
    d2 = QuasiRandomDesigner(space, seed=3)
    d2.load(state)
    cont_a = d1.suggest(5)
    cont_b = d2.suggest(5)
    for a, b in zip(cont_a, cont_b):
      assert a.parameters.as_dict() == b.parameters.as_dict()
    check_feasible(space, first + cont_a)

  def test_log_scale_mapping(self):
    space = vz.SearchSpace()
    space.root.add_float_param('lr', 1e-4, 1.0, scale_type=vz.ScaleType.LOG)
    designer = QuasiRandomDesigner(space, seed=0)
    lrs = [s.parameters.get_value('lr') for s in designer.suggest(256)]
    # Log-uniform: ~half the mass below 1e-2 (the geometric midpoint).
    frac_below = np.mean([lr < 1e-2 for lr in lrs])
    assert 0.35 < frac_below < 0.65


class TestGridSearchDesigner:

  def test_covers_grid_exactly(self):
    space = vz.SearchSpace()
    space.root.add_categorical_param('c', ['a', 'b'])
    space.root.add_discrete_param('d', [0.0, 1.0])
    designer = GridSearchDesigner(space)
    points = {tuple(sorted(s.parameters.as_dict().items()))
              for s in designer.suggest(4)}
    assert len(points) == 4

  def test_wraps_and_serializes(self):
    space = vz.SearchSpace()
    space.root.add_categorical_param('c', ['a', 'b'])
    designer = GridSearchDesigner(space)
    first4 = [s.parameters.get_value('c') for s in designer.suggest(4)]
    assert first4 == ['a', 'b', 'a', 'b']
    state = designer.dump()
    d2 = GridSearchDesigner(space)
    d2.load(state)
    assert d2.suggest(1)[0].parameters.get_value('c') == 'a'

  def test_shuffled_grid(self):
    space = vz.SearchSpace()
    space.root.add_int_param('i', 0, 9)
    designer = GridSearchDesigner(space, shuffle_seed=7)
    vals = [s.parameters.get_value('i') for s in designer.suggest(10)]
    assert sorted(vals) == list(range(10))
    assert vals != list(range(10))


class TestPolicies:

  def test_designer_policy_through_supporter(self):
    supporter = InRamPolicySupporter(problem())
    policy = DesignerPolicy(supporter,
                            lambda p: RandomDesigner(p.search_space, seed=1))
    trials = supporter.SuggestTrials(policy, 5)
    assert [t.id for t in trials] == [1, 2, 3, 4, 5]

  def test_partially_serializable_policy_resumes(self):
    prob = problem()
    supporter = InRamPolicySupporter(prob)
    policy = PartiallySerializableDesignerPolicy(
        prob, supporter, QuasiRandomDesigner.from_problem, ns_root='qr')
    t1 = supporter.SuggestTrials(policy, 3)
    t2 = supporter.SuggestTrials(policy, 3)
    # Fresh designer with the same seed: first 6 points == t1 + t2 (the
    # policy resumed the sequence rather than restarting it).
    fresh = QuasiRandomDesigner.from_problem(prob)
    expected = fresh.suggest(6)
    got = list(t1) + list(t2)
    for e, g in zip(expected, got):
      assert e.parameters.as_dict() == g.parameters.as_dict()

  def test_in_ram_supporter_best_trials(self):
    prob = problem()
    supporter = InRamPolicySupporter(prob)
    policy = DesignerPolicy(supporter,
                            lambda p: RandomDesigner(p.search_space, seed=2))
    trials = supporter.SuggestTrials(policy, 10)
    for i, t in enumerate(trials):
      t.complete(vz.Measurement(metrics={'obj': float(i)}))
    best = supporter.GetBestTrials(count=2)
    assert [t.final_measurement.metrics['obj'].value for t in best] == \
        [9.0, 8.0]
