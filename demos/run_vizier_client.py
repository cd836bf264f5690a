"""Runs a study against a Vizier server (or in-process).

Parity with the reference's demos/run_vizier_client.py.

Usage:
  python demos/run_vizier_client.py --endpoint localhost:28080 \
      --max_num_iterations 10
"""

import argparse
import logging
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import numpy as np

from vizier_amd import pyvizier as vz
from vizier_amd.service import clients


def evaluate(learning_rate: float, num_layers: int) -> float:
  return learning_rate * num_layers  # Toy objective.


def main() -> None:
  parser = argparse.ArgumentParser(description=__doc__)
  parser.add_argument('--endpoint', default=None,
                      help='host:port of a Vizier server; omit for '
                           'in-process.')
  parser.add_argument('--max_num_iterations', type=int, default=10)
  parser.add_argument('--algorithm', default='GP_UCB_PE')
  args = parser.parse_args()
  logging.basicConfig(level=logging.INFO)

  if args.endpoint:
    clients.environment_variables.server_endpoint = args.endpoint

  study_config = vz.StudyConfig(algorithm=args.algorithm)
  root = study_config.search_space.root
  root.add_float_param('learning_rate', 1e-4, 1e-2,
                       scale_type=vz.ScaleType.LOG)
  root.add_int_param('num_layers', 1, 5)
  study_config.metric_information.append(vz.MetricInformation(
      name='accuracy', goal=vz.ObjectiveMetricGoal.MAXIMIZE,
      min_value=0.0, max_value=1.0))

  study = clients.Study.from_study_config(study_config, owner='my_name',
                                          study_id='demo')
  for i in range(args.max_num_iterations):
    for trial in study.suggest(count=1):
      params = trial.parameters
      objective = evaluate(params['learning_rate'], params['num_layers'])
      trial.complete(vz.Measurement(metrics={'accuracy': objective}))
      print(f'Trial {trial.id}: {dict(params)} -> {objective:.5f}')

  optimal = list(study.optimal_trials().get())
  print('Optimal trial:', optimal[0].parameters.as_dict(),
        optimal[0].final_measurement.metrics['accuracy'].value)


if __name__ == '__main__':
  main()
