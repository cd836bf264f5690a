"""Demo: benchmark a designer on BBOB with the runner + analyzer stack.

Usage: python demos/run_benchmark.py [--designer random|cmaes|gp]
"""

import argparse
import sys

sys.path.insert(0, '.')

from vizier.benchmarks import (
    BenchmarkRunner,
    BenchmarkState,
    GenerateAndEvaluate,
)
from vizier.benchmarks import analyzers, experimenters


def main():
  parser = argparse.ArgumentParser()
  parser.add_argument('--designer', default='random',
                      choices=['random', 'cmaes', 'gp'])
  parser.add_argument('--trials', type=int, default=50)
  parser.add_argument('--repeats', type=int, default=3)
  args = parser.parse_args()

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
