"""Guards the bench.py driver contract (single-rank, tiny sizes).

The round driver invokes `python bench.py --gpus N --steps K --warmup
W` and parses ONE JSON line from rank 0 with an exact field set; this
test runs main() in-process at reduced N_TRIALS/MAX_EVALS and checks
every contract field survives refactors.
"""

import json
import sys

import pytest


@pytest.fixture()
def tiny_bench(monkeypatch):
  import bench
  monkeypatch.setattr(bench, 'N_TRIALS', 40)
  monkeypatch.setattr(bench, 'MAX_EVALS', 300)
  return bench


def _run(bench_mod, monkeypatch, capsys, argv):
  monkeypatch.setattr(sys, 'argv', ['bench.py'] + argv)
  bench_mod.main()
  lines = [ln for ln in capsys.readouterr().out.strip().splitlines()
           if ln.startswith('{')]
  assert len(lines) == 1, 'rank 0 must print exactly one JSON line'
  return json.loads(lines[-1])


def test_json_contract_fields(tiny_bench, monkeypatch, capsys):
  d = _run(tiny_bench, monkeypatch, capsys,
           ['--steps', '1', '--warmup', '0'])
  assert d['metric'] == 'suggest_wall_clock_ms_gp_bandit_20d_n1000'
  assert d['unit'] == 'ms'
  assert d['n_gpus'] == 1
  assert d['steps'] == 1 and d['warmup'] == 0
  assert d['higher_is_better'] is False
  assert d['scaling'] == 'weak'
  assert d['vs_baseline'] is None
  assert d['dtype'] == 'fp32'
  assert d['data'] == 'synthetic'
  assert d['value'] == d['ms_per_step'] > 0
  cfg = d['config']
  assert cfg['model'] == 'gp_bandit_matern52_ucb_eagle'
  assert cfg['dim'] == 20
  assert cfg['eagle_batch'] == 25
  assert 'parallelism' in cfg and cfg['parallelism'].startswith('dp1')
  assert cfg['ard'] == 'warm_lbfgs_12it_x2'
  assert 'global_batch' in cfg and 'seq_len' in cfg


def test_full_refit_flag_changes_ard_label(tiny_bench, monkeypatch,
                                           capsys):
  d = _run(tiny_bench, monkeypatch, capsys,
           ['--steps', '1', '--warmup', '0', '--full-refit'])
  assert d['config']['ard'] == 'lbfgs_50it_x5_restarts_cold'


def test_fp64_flag_reports_dtype(tiny_bench, monkeypatch, capsys):
  d = _run(tiny_bench, monkeypatch, capsys,
           ['--steps', '1', '--warmup', '0', '--fp64'])
  assert d['dtype'] == 'fp64'
