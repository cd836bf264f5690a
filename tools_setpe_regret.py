"""SetPE regret parity: batched GP-UCB-PE with/without the set
acquisition, same seeds/budget (VERDICT r1 #4 'done' criterion)."""
import json, sys, time
import numpy as np
sys.path.insert(0, '.')
from vizier_amd import pyvizier as vz
from vizier_amd._src.algorithms.core.abstractions import ActiveTrials, CompletedTrials
from vizier_amd._src.algorithms.designers.gp_ucb_pe import UCBPEConfig, VizierGPUCBPEBandit
from vizier_amd._src.benchmarks.experimenters.synthetic import bbob

DIM, BATCH, ROUNDS, SEEDS = 10, 4, 15, 3   # 60 trials per run


def run(set_pe: bool, fn, seed: int) -> float:
  p = vz.ProblemStatement()
  for i in range(DIM):
    p.search_space.root.add_float_param(f'x{i}', -5.0, 5.0)
  p.metric_information.append(vz.MetricInformation(
      name='obj', goal=vz.ObjectiveMetricGoal.MAXIMIZE))
  d = VizierGPUCBPEBandit(p, UCBPEConfig(
      max_evaluations=2000, ard_restarts=2, ard_max_iters=15,
      optimize_set_acquisition_for_exploration=set_pe,
      device='cpu'), seed=seed)
  shift = np.random.default_rng(500 + seed).uniform(-2, 2, DIM)
  best, uid = np.inf, 0
  for _ in range(ROUNDS):
    batch = d.suggest(BATCH)
    done = []
    for s in batch:
      uid += 1
      x = np.array([s.parameters.get_value(f'x{i}') for i in range(DIM)])
      v = fn(x - shift, seed=seed)
      best = min(best, v)
      t = s.to_trial(uid)
      t.complete(vz.Measurement(metrics={'obj': -v}))
      done.append(t)
    d.update(CompletedTrials(done), ActiveTrials())
  return float(best)


def main():
  out = {}
  for fname, fn in (('Sphere', bbob.Sphere), ('Rastrigin', bbob.Rastrigin),
                    ('SharpRidge', bbob.SharpRidge)):
    for mode in (False, True):
      key = f'{fname}/{"set_pe" if mode else "sequential_pe"}'
      vals = []
      for seed in range(SEEDS):
        t0 = time.time()
        vals.append(run(mode, fn, seed))
        print(f'{key} seed={seed} best@{BATCH*ROUNDS}={vals[-1]:.3f} '
              f'({time.time()-t0:.0f}s)', flush=True)
      out[key] = vals
  with open('profiles/setpe_regret_r2.json', 'w') as f:
    json.dump({'dim': DIM, 'batch': BATCH, 'rounds': ROUNDS,
               'best_value_minimized': out}, f, indent=2)
  print('wrote profiles/setpe_regret_r2.json')


if __name__ == '__main__':
  main()
