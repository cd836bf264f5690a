"""Demo: benchmark a designer on BBOB with the runner + analyzer stack.

Usage: python demos/run_benchmark.py [--designer random|cmaes|gp]
       python demos/run_benchmark.py --problem dtlz2   # multi-objective
"""

import argparse
import sys

sys.path.insert(0, '.')

import numpy as np

from vizier.benchmarks import (
    BenchmarkRunner,
    BenchmarkState,
    GenerateAndEvaluate,
)
from vizier.benchmarks import analyzers, experimenters


def run_multiobjective(args):
  """NSGA-II on the native DTLZ2 suite + hypervolume of the result."""
  from vizier_amd._src.algorithms.core.abstractions import (
      ActiveTrials,
      CompletedTrials,
  )
  from vizier_amd._src.algorithms.evolution.numpy_populations import (
      canonical_nsga2,
  )
  from vizier_amd._src.pyvizier.multimetric import ParetoFrontier

  factory = experimenters.DTLZExperimenterFactory(
      name='DTLZ2', dim=7, num_objectives=3)
  exp = factory()
  problem = exp.problem_statement()
  designer = canonical_nsga2(problem, population_size=40, seed=0)
  uid = 0
  for _ in range(args.trials):
    suggestions = designer.suggest(10)
    trials = []
    for s in suggestions:
      uid += 1
      trials.append(s.to_trial(uid))
    exp.evaluate(trials)
    designer.update(CompletedTrials(trials), ActiveTrials())
  # Objectives are MINIMIZE; hypervolume works on maximization values.
  ys = -designer.population.ys
  hv = ParetoFrontier(ys, origin=np.full(3, -3.0), seed=0).hypervolume()
  print(f'NSGA-II on DTLZ2 (7D, 3 objectives): final population '
        f'{len(designer.population)}, dominated hypervolume '
        f'{float(np.asarray(hv).reshape(-1)[-1]):.3f} '
        f'(origin (-3,-3,-3))')


def main():
  parser = argparse.ArgumentParser()
  parser.add_argument('--designer', default='random',
                      choices=['random', 'cmaes', 'gp'])
  parser.add_argument('--problem', default='sphere',
                      choices=['sphere', 'dtlz2'])
  parser.add_argument('--trials', type=int, default=50)
  parser.add_argument('--repeats', type=int, default=3)
  args = parser.parse_args()

  if args.problem == 'dtlz2':
    run_multiobjective(args)
    return

  factory = experimenters.BBOBExperimenterFactory(name='Sphere', dim=6)

  def designer_factory(problem, seed=0):
    if args.designer == 'cmaes':
      from vizier_amd._src.algorithms.designers.cmaes import CMAESDesigner
      return CMAESDesigner(problem, seed=seed)
    if args.designer == 'gp':
      from vizier_amd._src.algorithms.designers.gp_bandit import (
          GPBanditConfig, VizierGPBandit)
      return VizierGPBandit(problem, GPBanditConfig(
          max_evaluations=2000, ard_restarts=2, ard_max_iters=20))
    from vizier_amd._src.algorithms.designers.random import RandomDesigner
    return RandomDesigner(problem.search_space, seed=seed)

  states = []
  for rep in range(args.repeats):
    state = BenchmarkState.from_designer_factory(
        designer_factory, factory(), seed=rep)
    BenchmarkRunner([GenerateAndEvaluate(1)],
                    num_repeats=args.trials).run(state)
    states.append(state)

  curve = analyzers.BenchmarkStateAnalyzer.to_curve(states)
  final = curve.ys[:, -1]
  print(f'{args.designer} on Sphere-6D: best@{args.trials} over '
        f'{args.repeats} repeats = {final.tolist()}')


if __name__ == '__main__':
  main()
