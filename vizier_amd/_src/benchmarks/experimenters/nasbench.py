"""NAS-Bench-101 / NAS-Bench-201 experimenters (offline-gated).

Capability parity with
vizier/_src/benchmarks/experimenters/nasbench101_experimenter.py
(NASBench101Experimenter :45) and nasbench201_experimenter.py
(NASBench201Experimenter :45). Both take the benchmark API object by
injection (duck-typed), so the heavyweight datasets / packages
(nasbench, nats_bench — unavailable offline) are only needed when the
caller has them. `load_nasbench101` / `load_nasbench201` degrade
gracefully with an informative ImportError.

For offline tests and demos, `SyntheticNASBench101` /
`SyntheticNASBench201` implement the same query API with a
deterministic hash-based response surface, so every code path
(spec building, validity, metric fan-out) runs without the real data.
"""

from __future__ import annotations

import hashlib
from typing import Optional, Sequence

import numpy as np

from vizier_amd import pyvizier as vz
from vizier_amd._src.benchmarks.experimenters import experimenter

_NB101_OPS = ['conv3x3-bn-relu', 'conv1x1-bn-relu', 'maxpool3x3']
_NB201_OPS = ['none', 'skip_connect', 'nor_conv_1x1', 'nor_conv_3x3',
              'avg_pool_3x3']


def load_nasbench101(path: str):
  """Loads the real NAS-Bench-101 API; raises if the package is absent."""
  try:
    from nasbench import api  # pytype: disable=import-error
  except ImportError as e:
    raise ImportError(
        'NAS-Bench-101 requires the `nasbench` package and its '
        'nasbench_only108.tfrecord data file (not available offline). '
        'Install them, or use SyntheticNASBench101 for a data-free '
        'stand-in.') from e
  return api.NASBench(path)


def load_nasbench201(path: Optional[str] = None):
  """Loads the real NATS/NAS-Bench-201 topology API."""
  try:
    import nats_bench  # pytype: disable=import-error
  except ImportError as e:
    raise ImportError(
        'NAS-Bench-201 requires the `nats_bench` package and its data '
        'archive (not available offline). Install them, or use '
        'SyntheticNASBench201 for a data-free stand-in.') from e
  return nats_bench.create(path, 'tss', fast_mode=True)


def _hash01(key: str) -> float:
  h = hashlib.sha256(key.encode()).digest()
  return int.from_bytes(h[:8], 'little') / float(1 << 64)


class _ModelSpec:
  """Matrix+ops spec mirroring nasbench.api.ModelSpec's surface."""

  def __init__(self, matrix: np.ndarray, ops: Sequence[str]):
    self.matrix = matrix
    self.ops = list(ops)


class SyntheticNASBench101:
  """Deterministic data-free stand-in for nasbench.api.NASBench."""

  def is_valid(self, spec) -> bool:
    # Real constraint: <= 9 edges and output reachable from input.
    return int(spec.matrix.sum()) <= 9 and spec.matrix[0].any()

  def query(self, spec) -> dict:
    key = spec.matrix.tobytes().hex() + '|'.join(spec.ops)
    base = _hash01(key)
    return {
        'trainable_parameters': int(1e6 * (1 + base)),
        'training_time': 3600.0 * (0.5 + base),
        'train_accuracy': 0.5 + 0.5 * base,
        'validation_accuracy': 0.4 + 0.5 * base,
        'test_accuracy': 0.4 + 0.5 * base,
    }


class NASBench101Experimenter(experimenter.Experimenter):
  """7-vertex DAG topology (21 bools) + 5 op categoricals (:45-115)."""

  METRIC_NAMES = ['trainable_parameters', 'training_time',
                  'train_accuracy', 'validation_accuracy',
                  'test_accuracy']

  def __init__(self, nasbench):
    self._nasbench = nasbench
    self._num_vertices = 7
    self._op_spots = self._num_vertices - 2

  def problem_statement(self) -> vz.ProblemStatement:
    problem = vz.ProblemStatement()
    root = problem.search_space.root
    for y in range(self._num_vertices):
      for x in range(self._num_vertices):
        if y > x:
          root.add_bool_param(f'{x}_{y}')
    for i in range(self._op_spots):
      root.add_categorical_param(f'ops_{i}', _NB101_OPS)
    problem.metric_information.append(vz.MetricInformation(
        name='validation_accuracy',
        goal=vz.ObjectiveMetricGoal.MAXIMIZE))
    return problem

  def _trial_to_model_spec(self, trial: vz.Trial) -> _ModelSpec:
    n = self._num_vertices
    matrix = np.zeros((n, n), dtype=int)
    for y in range(n):
      for x in range(n):
        if y > x:
          v = trial.parameters.get_value(f'{x}_{y}')
          matrix[x][y] = int(str(v).lower() == 'true')
    ops = (['input'] +
           [trial.parameters.get_value(f'ops_{i}')
            for i in range(self._op_spots)] + ['output'])
    return _ModelSpec(matrix, ops)

  def evaluate(self, suggestions: Sequence[vz.Trial]) -> None:
    for trial in suggestions:
      spec = self._trial_to_model_spec(trial)
      if self._nasbench.is_valid(spec):
        results = self._nasbench.query(spec)
        trial.complete(vz.Measurement(
            metrics={k: results[k] for k in self.METRIC_NAMES}))
      else:
        trial.complete(vz.Measurement(),
                       infeasibility_reason='Not in search space.')


def model_tss_spec(ops: Sequence[str], num_nodes: int) -> str:
  """ops list -> NATS topology string (nasbench201_experimenter.py:33)."""
  nodes, k = [], 0
  for i in range(1, num_nodes):
    parts = []
    for j in range(i):
      parts.append(f'{ops[k]}~{j}')
      k += 1
    nodes.append('|' + '|'.join(parts) + '|')
  return '+'.join(nodes)


class SyntheticNASBench201:
  """Deterministic stand-in for the NATS topology API."""

  def query_index_by_arch(self, arch: str) -> int:
    return int(_hash01(arch) * (1 << 30))

  def get_more_info(self, index: int, dataset: str, hp: str = '12'
                    ) -> dict:
    base = _hash01(f'{index}|{dataset}|{hp}')
    return {'valid-accuracy': 40.0 + 50.0 * base,
            'test-accuracy': 40.0 + 50.0 * base,
            'train-all-time': 100.0 * (1 + base)}


class NASBench201Experimenter(experimenter.Experimenter):
  """Topology search space: 6 op slots over 4 nodes (:45-100)."""

  def __init__(self, nasbench, dataset_str: str = 'cifar10',
               validation_set_reporting_epoch: int = 12):
    self._nasbench = nasbench
    self._dataset_str = dataset_str
    self._epoch = validation_set_reporting_epoch
    self._num_nodes = 4
    self._op_spots = 6

  def problem_statement(self) -> vz.ProblemStatement:
    problem = vz.ProblemStatement()
    for i in range(self._op_spots):
      problem.search_space.root.add_categorical_param(
          f'op_{i}', _NB201_OPS)
    problem.metric_information.append(vz.MetricInformation(
        name='valid-accuracy', goal=vz.ObjectiveMetricGoal.MAXIMIZE))
    return problem

  def _trial_to_arch(self, trial: vz.Trial) -> str:
    ops = [trial.parameters.get_value(f'op_{i}')
           for i in range(self._op_spots)]
    return model_tss_spec(ops, self._num_nodes)

  def evaluate(self, suggestions: Sequence[vz.Trial]) -> None:
    for trial in suggestions:
      arch = self._trial_to_arch(trial)
      index = self._nasbench.query_index_by_arch(arch)
      info = self._nasbench.get_more_info(index, self._dataset_str,
                                          hp=str(self._epoch))
      trial.complete(vz.Measurement(metrics={
          'valid-accuracy': info['valid-accuracy'],
          'test-accuracy': info.get('test-accuracy', 0.0),
      }))
