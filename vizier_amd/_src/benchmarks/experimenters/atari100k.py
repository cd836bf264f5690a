"""Atari100k RL tuning experimenter (heavyweight, dependency-gated).

Capability parity with
vizier/_src/benchmarks/experimenters/atari100k_experimenter.py
(default_search_space :77, Atari100kExperimenter :111). The real
benchmark trains a Dopamine Rainbow-family agent per trial (gin-config
driven); Dopamine/JAX/gin are unavailable offline, so:

- `default_search_space()` — the full 14-parameter tuning space — is
  always available (pure pyvizier);
- `Atari100kExperimenter` exposes the same constructor/problem
  surface; `evaluate` raises ImportError with install guidance unless
  a `runner_factory` is injected (tests inject a fake runner, real
  users inject a Dopamine-backed one).
"""

from __future__ import annotations

from typing import Callable, Dict, Optional, Sequence, Union

from vizier_amd import pyvizier as vz
from vizier_amd._src.benchmarks.experimenters import experimenter

GinParameterType = Union[bool, int, float, str]

_AGENTS = ('DER', 'DrQ', 'DrQ_eps', 'OTRainbow')


def default_search_space() -> vz.SearchSpace:
  """The 14-parameter agent tuning space (atari100k_experimenter.py:77)."""
  ss = vz.SearchSpace()
  root = ss.root
  root.add_float_param('JaxDQNAgent.gamma', 0.7, 1.0,
                       scale_type=vz.ScaleType.REVERSE_LOG)
  root.add_int_param('JaxDQNAgent.update_horizon', 1, 20)
  root.add_int_param('JaxDQNAgent.update_period', 1, 10)
  root.add_int_param('JaxDQNAgent.target_update_period', 1, 10000)
  root.add_int_param('JaxDQNAgent.min_replay_history', 100, 100000)
  root.add_float_param('JaxDQNAgent.epsilon_train', 1e-5, 0.5,
                       scale_type=vz.ScaleType.LOG)
  root.add_int_param('JaxDQNAgent.epsilon_decay_period', 1000, 10000)
  root.add_bool_param('JaxFullRainbowAgent.noisy')
  root.add_bool_param('JaxFullRainbowAgent.dueling')
  root.add_bool_param('JaxFullRainbowAgent.double_dqn')
  root.add_int_param('JaxFullRainbowAgent.num_atoms', 1, 100)
  root.add_bool_param('Atari100kRainbowAgent.data_augmentation')
  root.add_float_param('create_optimizer.learning_rate', 1e-6, 1e-2,
                       scale_type=vz.ScaleType.LOG)
  root.add_float_param('create_optimizer.eps', 1e-9, 1e-2,
                       scale_type=vz.ScaleType.LOG)
  return ss


class Atari100kExperimenter(experimenter.Experimenter):
  """Trains an agent per trial through an injected runner factory.

  runner_factory(game_name, agent_name, bindings) -> runner with a
  `run_trial()` returning {'eval_average_return': [floats per epoch]}.
  """

  def __init__(self, game_name: str = 'Pong', agent_name: str = 'DER',
               initial_gin_bindings: Optional[
                   Dict[str, GinParameterType]] = None,
               runner_factory: Optional[Callable] = None):
    if agent_name not in _AGENTS:
      raise ValueError(f'agent_name must be one of {_AGENTS}')
    self._game_name = game_name
    self._agent_name = agent_name
    self._initial_gin_bindings = dict(initial_gin_bindings or {})
    self._runner_factory = runner_factory

  def problem_statement(self) -> vz.ProblemStatement:
    problem = vz.ProblemStatement(search_space=default_search_space())
    problem.metric_information.append(vz.MetricInformation(
        name='eval_average_return',
        goal=vz.ObjectiveMetricGoal.MAXIMIZE))
    return problem

  def evaluate(self, suggestions: Sequence[vz.Trial]) -> None:
    if self._runner_factory is None:
      raise ImportError(
          'Atari100k evaluation needs the Dopamine RL stack '
          '(dopamine-rl, gin-config, an Atari ROM set), which is not '
          'available offline. Inject runner_factory=... to supply a '
          'training backend.')
    for trial in suggestions:
      bindings = dict(self._initial_gin_bindings)
      bindings.update(
          {name: trial.parameters.get_value(name)
           for name in trial.parameters})
      runner = self._runner_factory(self._game_name, self._agent_name,
                                    bindings)
      stats = runner.run_trial()
      returns = list(stats['eval_average_return'])
      for step, value in enumerate(returns[:-1]):
        trial.measurements.append(vz.Measurement(
            metrics={'eval_average_return': float(value)},
            steps=step + 1))
      trial.complete(vz.Measurement(
          metrics={'eval_average_return': float(returns[-1])},
          steps=len(returns)))
