"""GP-UCB-PE designer tests (CPU)."""

import numpy as np
import pytest
import torch

from vizier_amd import pyvizier as vz
from vizier_amd._src.algorithms.core.abstractions import (
    ActiveTrials,
    CompletedTrials,
)
from vizier_amd._src.algorithms.designers.gp_ucb_pe import (
    UCBPEConfig,
    VizierGPUCBPEBandit,
)


def make_problem(dim=3) -> vz.ProblemStatement:
  problem = vz.ProblemStatement()
  for i in range(dim):
    problem.search_space.root.add_float_param(f'x{i}', 0.0, 1.0)
  problem.metric_information.append(
      vz.MetricInformation(name='obj', goal=vz.ObjectiveMetricGoal.MAXIMIZE))
  return problem


def cfg(**kw) -> UCBPEConfig:
  base = dict(max_evaluations=800, ard_restarts=2, ard_max_iters=15,
              device='cpu')
  base.update(kw)
  return UCBPEConfig(**base)


def evaluate(s) -> float:
  x = np.array([s.parameters.get_value(f'x{i}') for i in range(3)])
  return float(-((x - 0.6) ** 2).sum())


class TestGPUCBPE:

  def test_first_batch_mixes_ucb_and_pe(self):
    designer = VizierGPUCBPEBandit(make_problem(), cfg(), seed=1)
    trials = []
    for uid in range(1, 6):
      s = vz.TrialSuggestion(
          {f'x{i}': float(v) for i, v in
           enumerate(np.random.default_rng(uid).uniform(0, 1, 3))})
      t = s.to_trial(uid)
      t.complete(vz.Measurement(metrics={'obj': evaluate(s)}))
      trials.append(t)
    designer.update(CompletedTrials(trials), ActiveTrials())
    batch = designer.suggest(3)
    assert len(batch) == 3
    kinds = [s.metadata.abs_ns(('gp_ucb_pe',))['acquisition']
             for s in batch]
    # First member exploits (modulo the 10% PE overwrite with this seed),
    # later members explore.
    assert 'pe' in kinds
    # Batch members are distinct points (PE avoids duplicates).
    points = {tuple(s.parameters.as_dict().values()) for s in batch}
    assert len(points) == 3

  def test_convergence_beats_random(self):
    designer = VizierGPUCBPEBandit(make_problem(), cfg(), seed=2)
    uid, best = 0, -np.inf
    for _ in range(10):
      for s in designer.suggest(1):
        uid += 1
        val = evaluate(s)
        best = max(best, val)
        t = s.to_trial(uid)
        t.complete(vz.Measurement(metrics={'obj': val}))
        designer.update(CompletedTrials([t]), ActiveTrials())
    rng = np.random.default_rng(2)
    random_best = max(float(-((rng.uniform(0, 1, 3) - 0.6) ** 2).sum())
                      for _ in range(10))
    assert best > random_best
    assert best > -0.05

  def test_pending_trials_fuel_pure_exploration(self):
    designer = VizierGPUCBPEBandit(make_problem(), cfg(), seed=3)
    trials = []
    for uid in range(1, 5):
      s = vz.TrialSuggestion({f'x{i}': 0.1 * uid for i in range(3)})
      t = s.to_trial(uid)
      t.complete(vz.Measurement(metrics={'obj': evaluate(s)}))
      trials.append(t)
    active = [vz.TrialSuggestion(
        {f'x{i}': 0.9 for i in range(3)}).to_trial(99)]
    designer.update(CompletedTrials(trials), ActiveTrials(active))
    out = designer.suggest(1)
    assert len(out) == 1

  def test_set_pe_batch_diversity(self):
    # SetPE (logdet joint acquisition, reference gp_ucb_pe.py:510):
    # exploration members of a batch must DECORRELATE — pairwise
    # distances of the jointly-optimized set are bounded away from 0,
    # and the suggest flow labels them 'set_pe'.
    designer = VizierGPUCBPEBandit(
        make_problem(), cfg(optimize_set_acquisition_for_exploration=True,
                            max_evaluations=1500), seed=5)
    trials = []
    rng = np.random.default_rng(0)
    for uid in range(1, 9):
      s = vz.TrialSuggestion(
          {f'x{i}': float(v) for i, v in enumerate(rng.uniform(0, 1, 3))})
      t = s.to_trial(uid)
      t.complete(vz.Measurement(metrics={'obj': evaluate(s)}))
      trials.append(t)
    designer.update(CompletedTrials(trials), ActiveTrials())
    batch = designer.suggest(4)
    assert len(batch) == 4
    kinds = [s.metadata.abs_ns(('gp_ucb_pe',))['acquisition']
             for s in batch]
    # One UCB exploit (new data arrived), rest jointly explored.
    assert kinds.count('set_pe') == 3
    pts = np.array([[s.parameters.get_value(f'x{i}') for i in range(3)]
                    for s in batch if 'set_pe' in
                    s.metadata.abs_ns(('gp_ucb_pe',))['acquisition']])
    dists = [np.abs(pts[a] - pts[b]).max()
             for a in range(len(pts)) for b in range(a + 1, len(pts))]
    # logdet of the joint covariance collapses to -inf for duplicate
    # points, so the optimized set must be spread out.
    assert min(dists) > 1e-3

  def test_set_pe_no_new_trials_all_set(self):
    designer = VizierGPUCBPEBandit(
        make_problem(), cfg(optimize_set_acquisition_for_exploration=True,
                            max_evaluations=800), seed=6)
    trials = []
    rng = np.random.default_rng(1)
    for uid in range(1, 7):
      s = vz.TrialSuggestion(
          {f'x{i}': float(v) for i, v in enumerate(rng.uniform(0, 1, 3))})
      t = s.to_trial(uid)
      t.complete(vz.Measurement(metrics={'obj': evaluate(s)}))
      trials.append(t)
    designer.update(CompletedTrials(trials), ActiveTrials())
    designer.suggest(1)   # consumes the has-new-trials credit
    batch = designer.suggest(3)
    kinds = [s.metadata.abs_ns(('gp_ucb_pe',))['acquisition']
             for s in batch]
    # No new completed trials since the last suggest: the whole batch
    # comes from the set acquisition (gp_ucb_pe.py:1423-1431).
    assert kinds == ['set_pe'] * 3

  def test_seed_phase(self):
    designer = VizierGPUCBPEBandit(make_problem(), cfg(), seed=4)
    first = designer.suggest(2)
    assert len(first) == 2
    assert first[0].parameters.get_value('x0') == pytest.approx(0.5)


class TestUCBPELinearKernel:

  def test_mixes_linear_kernel_both_phases(self):
    import numpy as np
    from vizier_amd import pyvizier as vz
    from vizier_amd._src.algorithms.core.abstractions import (
        ActiveTrials, CompletedTrials)
    from vizier_amd._src.algorithms.designers.gp_ucb_pe import (
        UCBPEConfig, VizierGPUCBPEBandit)
    from vizier_amd._src.gp.linear_matern import LinearMaternPosterior
    p = vz.ProblemStatement()
    for i in range(2):
      p.search_space.root.add_float_param(f'x{i}', -1.0, 1.0)
    p.metric_information.append(vz.MetricInformation(
        name='m', goal=vz.ObjectiveMetricGoal.MAXIMIZE))
    rng = np.random.default_rng(0)
    trials = []
    for uid in range(1, 14):
      params = {f'x{i}': float(rng.uniform(-1, 1)) for i in range(2)}
      t = vz.Trial(params, id=uid)
      x = np.array(list(params.values()))
      t.complete(vz.Measurement(metrics={'m': float(x.sum())}))
      trials.append(t)
    d = VizierGPUCBPEBandit(p, UCBPEConfig(
        max_evaluations=300, ard_restarts=2, ard_max_iters=6,
        mixes_linear_kernel=True))
    d.update(CompletedTrials(trials), ActiveTrials())
    # count=3: UCB phase (combined-kernel scoring) + PE fill (combined-
    # kernel hallucinated variance posterior).
    assert len(d.suggest(3)) == 3
    assert isinstance(d._posterior, LinearMaternPosterior)
    # Warm refit through the linear raw layout.
    t2 = vz.Trial({'x0': 0.2, 'x1': 0.2}, id=99)
    t2.complete(vz.Measurement(metrics={'m': 0.4}))
    d.update(CompletedTrials([t2]), ActiveTrials())
    assert len(d.suggest(1)) == 1


class TestUCBPEMultimetric:

  def _problem(self):
    import numpy as np
    from vizier_amd import pyvizier as vz
    p = vz.ProblemStatement()
    for i in range(2):
      p.search_space.root.add_float_param(f'x{i}', -1.0, 1.0)
    for name in ('m1', 'm2'):
      p.metric_information.append(vz.MetricInformation(
          name=name, goal=vz.ObjectiveMetricGoal.MAXIMIZE))
    return p

  def test_default_algorithm_serves_mo_studies(self):
    import numpy as np
    from vizier_amd import pyvizier as vz
    from vizier_amd._src.algorithms.core.abstractions import (
        ActiveTrials, CompletedTrials)
    from vizier_amd._src.algorithms.designers.gp_ucb_pe import (
        UCBPEConfig, VizierGPUCBPEBandit)
    p = self._problem()
    rng = np.random.default_rng(0)
    trials = []
    for uid in range(1, 14):
      params = {f'x{i}': float(rng.uniform(-1, 1)) for i in range(2)}
      t = vz.Trial(params, id=uid)
      x = np.array(list(params.values()))
      t.complete(vz.Measurement(metrics={'m1': float(x.sum()),
                                         'm2': float(-(x**2).sum())}))
      trials.append(t)
    d = VizierGPUCBPEBandit(p, UCBPEConfig(
        max_evaluations=300, ard_restarts=2, ard_max_iters=6,
        num_scalarizations=50))
    d.update(CompletedTrials(trials), ActiveTrials())
    s = d.suggest(3)
    assert len(s) == 3
    assert d._mo_posteriors is not None and len(d._mo_posteriors) == 2
    # Both phases appear across a batch (first=UCB, fills=PE by
    # construction when there are fresh trials and no actives).
    kinds = {x.metadata.ns('gp_ucb_pe')['acquisition'] for x in s}
    assert 'pe' in kinds and 'ucb' in kinds
    # Warm refit across the per-metric fits.
    t2 = vz.Trial({'x0': 0.2, 'x1': 0.1}, id=99)
    t2.complete(vz.Measurement(metrics={'m1': 0.3, 'm2': -0.05}))
    d.update(CompletedTrials([t2]), ActiveTrials())
    assert len(d.suggest(1)) == 1

  def test_mo_prior_seeding_with_many_trials(self):
    # Regression: Eagle prior seeding takes SCALAR rewards; with
    # multi-metric labels and enough trials to overflow the firefly
    # pool the dedup comparison used to see (M,)-shaped rewards.
    import numpy as np
    from vizier_amd import pyvizier as vz
    from vizier_amd._src.algorithms.core.abstractions import (
        ActiveTrials, CompletedTrials)
    from vizier_amd._src.algorithms.designers.gp_ucb_pe import (
        UCBPEConfig, VizierGPUCBPEBandit)
    p = self._problem()
    rng = np.random.default_rng(1)
    trials = []
    for uid in range(1, 41):
      params = {f'x{i}': float(rng.uniform(-1, 1)) for i in range(2)}
      t = vz.Trial(params, id=uid)
      x = np.array(list(params.values()))
      t.complete(vz.Measurement(metrics={'m1': float(x.sum()),
                                         'm2': float(-(x**2).sum())}))
      trials.append(t)
    d = VizierGPUCBPEBandit(p, UCBPEConfig(
        max_evaluations=200, ard_restarts=2, ard_max_iters=5,
        num_scalarizations=30))
    d.update(CompletedTrials(trials), ActiveTrials())
    assert len(d.suggest(2)) == 2

  def test_mo_multitask_surrogate(self):
    import numpy as np
    from vizier_amd import pyvizier as vz
    from vizier_amd._src.algorithms.core.abstractions import (
        ActiveTrials, CompletedTrials)
    from vizier_amd._src.algorithms.designers.gp_ucb_pe import (
        UCBPEConfig, VizierGPUCBPEBandit)
    p = self._problem()
    rng = np.random.default_rng(2)
    trials = []
    for uid in range(1, 14):
      params = {f'x{i}': float(rng.uniform(-1, 1)) for i in range(2)}
      t = vz.Trial(params, id=uid)
      x = np.array(list(params.values()))
      t.complete(vz.Measurement(metrics={'m1': float(x.sum()),
                                         'm2': float(-(x**2).sum())}))
      trials.append(t)
    d = VizierGPUCBPEBandit(p, UCBPEConfig(
        max_evaluations=200, ard_restarts=2, ard_max_iters=5,
        num_scalarizations=30, multitask_type='separable'))
    d.update(CompletedTrials(trials), ActiveTrials())
    assert len(d.suggest(3)) == 3  # UCB + PE through the joint GP
    t2 = vz.Trial({'x0': 0.1, 'x1': 0.1}, id=99)
    t2.complete(vz.Measurement(metrics={'m1': 0.2, 'm2': -0.02}))
    d.update(CompletedTrials([t2]), ActiveTrials())
    assert len(d.suggest(1)) == 1  # joint warm refit


class TestFusedPathGuards:
  """The HIP fused scorers must NOT engage on CPU (torch fallback)."""

  def test_pe_phase_runs_on_cpu_with_pending(self):
    designer = VizierGPUCBPEBandit(make_problem(), cfg(), seed=8)
    trials = []
    rng = np.random.default_rng(2)
    for uid in range(1, 7):
      s = vz.TrialSuggestion(
          {f'x{i}': float(v) for i, v in enumerate(rng.uniform(0, 1, 3))})
      t = s.to_trial(uid)
      t.complete(vz.Measurement(metrics={'obj': evaluate(s)}))
      trials.append(t)
    active = [vz.TrialSuggestion(
        {f'x{i}': 0.9 for i in range(3)}).to_trial(99)]
    designer.update(CompletedTrials(trials), ActiveTrials(active))
    out = designer.suggest(2)
    assert len(out) == 2
    # _fusable is False on CPU tensors.
    assert not designer._fusable(designer._posterior)

  def test_set_pe_with_pending_trials(self):
    designer = VizierGPUCBPEBandit(
        make_problem(), cfg(optimize_set_acquisition_for_exploration=True,
                            max_evaluations=600), seed=9)
    trials = []
    rng = np.random.default_rng(3)
    for uid in range(1, 7):
      s = vz.TrialSuggestion(
          {f'x{i}': float(v) for i, v in enumerate(rng.uniform(0, 1, 3))})
      t = s.to_trial(uid)
      t.complete(vz.Measurement(metrics={'obj': evaluate(s)}))
      trials.append(t)
    active = [vz.TrialSuggestion(
        {f'x{i}': 0.8 for i in range(3)}).to_trial(50)]
    designer.update(CompletedTrials(trials), ActiveTrials(active))
    batch = designer.suggest(3)
    kinds = [s.metadata.abs_ns(('gp_ucb_pe',))['acquisition']
             for s in batch]
    # Pending trials condition the set covariance; batch still mixes
    # one UCB (new data) + jointly-optimized set members.
    assert kinds[0] == 'ucb'
    assert kinds[1:] == ['set_pe', 'set_pe']
