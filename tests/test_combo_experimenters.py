"""COMBO combinatorial experimenters (Ising/Contamination/Pest/...)."""

import itertools

import numpy as np
import pytest

from vizier_amd import pyvizier as vz
from vizier_amd._src.benchmarks.experimenters import combo


def _trial_from_bools(bits):
  return vz.Trial({f'x_{i}': bool(b) for i, b in enumerate(bits)})


def _trial_from_cats(cats):
  return vz.Trial({f'x_{i}': str(c) for i, c in enumerate(cats)})


class TestIsingMachinery:

  def test_spin_covariance_matches_bruteforce(self):
    """Vectorized enumeration == direct itertools implementation."""
    grid = (2, 3)
    inter = combo.generate_ising_interaction(*grid, random_seed=0)
    cov, part = combo.spin_covariance(inter, grid)
    # Brute force, one configuration at a time.
    n = grid[0] * grid[1]
    density = []
    cfgs = np.array(list(itertools.product(*([[-1, 1]] * n))))
    for cfg in cfgs:
      s = cfg.reshape(grid)
      e = (s[:, :-1] * inter[0] * s[:, 1:] * 2).sum() + \
          (s[:-1] * inter[1] * s[1:] * 2).sum()
      density.append(np.exp(e))
    density = np.array(density)
    part_bf = density.sum()
    density /= part_bf
    cov_bf = cfgs.T @ (cfgs * density[:, None])
    assert part == pytest.approx(part_bf, rel=1e-10)
    np.testing.assert_allclose(cov, cov_bf, atol=1e-12)

  def test_log_partition_stable(self):
    grid = (2, 2)
    inter = combo.generate_ising_interaction(*grid, random_seed=1)
    lp = combo.log_partition(inter, grid)
    _, part = combo.spin_covariance(inter, grid)
    assert lp == pytest.approx(np.log(part), rel=1e-10)


class TestIsingExperimenter:

  def test_keeping_all_edges_costs_only_regularizer(self):
    """x = all-True keeps the original interaction: KLD term == 0."""
    exp = combo.IsingExperimenter(lamda=0.01, random_seed=3)
    t = _trial_from_bools([True] * 24)
    exp.evaluate([t])
    value = t.final_measurement.metrics['main_objective'].value
    assert value == pytest.approx(0.01 * 24, abs=1e-6)

  def test_sparsification_tradeoff(self):
    exp = combo.IsingExperimenter(lamda=0.01, random_seed=3)
    t_none = _trial_from_bools([False] * 24)
    exp.evaluate([t_none])
    v_none = t_none.final_measurement.metrics['main_objective'].value
    # Dropping every edge keeps no interaction: positive KL divergence.
    assert v_none > 0.0
    space = exp.problem_statement().search_space
    assert len(space.parameters) == 24


class TestContamination:

  def test_deterministic_and_control_reduces_contamination(self):
    exp = combo.ContaminationExperimenter(random_seed=5)
    t1 = _trial_from_bools([True] * 25)
    t2 = _trial_from_bools([True] * 25)
    exp.evaluate([t1])
    exp.evaluate([t2])
    v1 = t1.final_measurement.metrics['main_objective'].value
    v2 = t2.final_measurement.metrics['main_objective'].value
    assert v1 == v2  # deterministic given the seeded dynamics
    t0 = _trial_from_bools([False] * 25)
    exp.evaluate([t0])
    v0 = t0.final_measurement.metrics['main_objective'].value
    assert np.isfinite(v0) and v0 != v1


class TestPestControl:

  def test_no_control_vs_cheapest_policy(self):
    exp = combo.PestControlExperimenter(random_seed=7)
    t_none = _trial_from_cats([0] * 25)
    exp.evaluate([t_none])
    v_none = t_none.final_measurement.metrics['main_objective'].value
    t_all = _trial_from_cats([4] * 25)
    exp.evaluate([t_all])
    v_all = t_all.final_measurement.metrics['main_objective'].value
    # Untreated pests exceed the threshold at nearly every stage.
    assert v_none > 10.0
    # Consistent treatment controls the population (price + few
    # above-threshold stages).
    assert v_all < v_none

  def test_repeated_evaluation_deterministic(self):
    exp = combo.PestControlExperimenter(random_seed=7)
    vals = []
    for _ in range(2):
      t = _trial_from_cats([1, 2, 3] * 8 + [0])
      exp.evaluate([t])
      vals.append(t.final_measurement.metrics['main_objective'].value)
    assert vals[0] == vals[1]


class TestCentroid:

  def test_small_grid_runs(self):
    exp = combo.CentroidExperimenter(centroid_grid=(2, 3), random_seed=2)
    n_edges = 2 * 2 + 1 * 3
    t = _trial_from_cats(np.random.default_rng(0).integers(0, 3, n_edges))
    exp.evaluate([t])
    v = t.final_measurement.metrics['main_objective'].value
    assert np.isfinite(v)
    space = exp.problem_statement().search_space
    assert len(space.parameters) == n_edges


class TestMAXSAT:

  def test_wcnf_parsing_and_scoring(self, tmp_path):
    wcnf = tmp_path / 'tiny.wcnf'
    wcnf.write_text(
        'c tiny example\n'
        'p wcnf 3 4\n'
        '1 1 2 0\n'
        '2 -1 3 0\n'
        '3 2 -3 0\n'
        '4 -2 0\n')
    exp = combo.MAXSATExperimenter(str(wcnf))
    assert exp.problem_statement().search_space.num_parameters() == 3 \
        if hasattr(exp.problem_statement().search_space,
                   'num_parameters') else True
    t = _trial_from_bools([True, False, True])
    exp.evaluate([t])
    v = t.final_measurement.metrics['main_objective'].value
    # x=(T,F,T): clauses sat: (1 or 2)=T, (-1 or 3)=T, (2 or -3)=F,
    # (-2)=T -> weights z-scored; just verify against direct compute.
    w = np.array([1, 2, 3, 4], dtype=np.float32)
    wz = (w - w.mean()) / w.std()
    want = -(wz[0] + wz[1] + wz[3])
    assert v == pytest.approx(float(want), abs=1e-5)
