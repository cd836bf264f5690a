"""Problem statement: metrics configuration + search space.

MI355X-native rewrite with the capabilities of
vizier/_src/pyvizier/shared/base_study_config.py (ObjectiveMetricGoal :55,
MetricType :71, MetricInformation :92, MetricsConfig :222,
ProblemStatement :306).
"""

from __future__ import annotations

import copy
import enum
import math
from typing import Callable, Iterable, Iterator, List, Optional, Type, TypeVar

from vizier_amd._src.pyvizier.common import Metadata
from vizier_amd._src.pyvizier.parameter_config import SearchSpace

_T = TypeVar('_T', bound='ProblemStatement')


class ObjectiveMetricGoal(enum.IntEnum):
  MAXIMIZE = 1
  MINIMIZE = 2

  @property
  def is_maximize(self) -> bool:
    return self == ObjectiveMetricGoal.MAXIMIZE

  @property
  def is_minimize(self) -> bool:
    return self == ObjectiveMetricGoal.MINIMIZE


class MetricType(enum.Enum):
  OBJECTIVE = 'OBJECTIVE'
  SAFETY = 'SAFETY'

  @property
  def is_safety(self) -> bool:
    return self == MetricType.SAFETY

  @property
  def is_objective(self) -> bool:
    return self == MetricType.OBJECTIVE


class MetricInformation:
  """Optimization configuration for one metric."""

  def __init__(self, name: str = '', *,
               goal: ObjectiveMetricGoal,
               safety_threshold: Optional[float] = None,
               safety_std_threshold: Optional[float] = None,
               desired_min_safe_trials_fraction: Optional[float] = None,
               min_value: Optional[float] = None,
               max_value: Optional[float] = None):
    self.name = name
    self.goal = ObjectiveMetricGoal(goal)
    self.safety_threshold = (float(safety_threshold)
                             if safety_threshold is not None else None)
    self.safety_std_threshold = (float(safety_std_threshold)
                                 if safety_std_threshold is not None else None)
    if desired_min_safe_trials_fraction is not None:
      f = float(desired_min_safe_trials_fraction)
      if not 0.0 <= f <= 1.0:
        raise ValueError('desired_min_safe_trials_fraction must be in [0,1]')
      self.desired_min_safe_trials_fraction = f
    else:
      self.desired_min_safe_trials_fraction = None
    self._min_value = float(min_value) if min_value is not None else -math.inf
    self._max_value = float(max_value) if max_value is not None else math.inf
    if self._min_value > self._max_value:
      raise ValueError(f'min_value {self._min_value} > max_value '
                       f'{self._max_value} for metric {name!r}')

  @property
  def min_value(self) -> float:
    return self._min_value

  @min_value.setter
  def min_value(self, v: Optional[float]):
    v = float(v) if v is not None else -math.inf
    if v > self._max_value:
      raise ValueError('min_value > max_value')
    self._min_value = v

  @property
  def max_value(self) -> float:
    return self._max_value

  @max_value.setter
  def max_value(self, v: Optional[float]):
    v = float(v) if v is not None else math.inf
    if v < self._min_value:
      raise ValueError('max_value < min_value')
    self._max_value = v

  def min_value_or(self, default_value_fn: Callable[[], float]) -> float:
    return self._min_value if math.isfinite(self._min_value) \
        else default_value_fn()

  def max_value_or(self, default_value_fn: Callable[[], float]) -> float:
    return self._max_value if math.isfinite(self._max_value) \
        else default_value_fn()

  @property
  def range(self) -> float:
    return self._max_value - self._min_value

  @property
  def type(self) -> MetricType:
    return (MetricType.SAFETY if self.safety_threshold is not None
            else MetricType.OBJECTIVE)

  def flip_goal(self) -> 'MetricInformation':
    self.goal = (ObjectiveMetricGoal.MINIMIZE if self.goal.is_maximize
                 else ObjectiveMetricGoal.MAXIMIZE)
    return self

  def __eq__(self, other) -> bool:
    if not isinstance(other, MetricInformation):
      return NotImplemented
    return (self.name == other.name and self.goal == other.goal and
            self.safety_threshold == other.safety_threshold and
            self.desired_min_safe_trials_fraction
            == other.desired_min_safe_trials_fraction and
            self._min_value == other._min_value and
            self._max_value == other._max_value)

  def __repr__(self) -> str:
    return (f'MetricInformation(name={self.name!r}, goal={self.goal.name}'
            + (f', safety_threshold={self.safety_threshold}'
               if self.safety_threshold is not None else '') + ')')


class MetricsConfig:
  """Ordered collection of MetricInformation with unique names."""

  def __init__(self, metrics: Iterable[MetricInformation] = ()):
    self._metrics: List[MetricInformation] = list(metrics)
    self._assert_unique()

  def _assert_unique(self):
    names = [m.name for m in self._metrics]
    if len(set(names)) != len(names):
      raise ValueError(f'Duplicate metric names: {names}')

  def item(self) -> MetricInformation:
    if len(self._metrics) != 1:
      raise ValueError(
          f'Expected exactly one metric; got {len(self._metrics)}')
    return self._metrics[0]

  def __iter__(self) -> Iterator[MetricInformation]:
    return iter(self._metrics)

  def __contains__(self, x) -> bool:
    return x in self._metrics

  def __len__(self) -> int:
    return len(self._metrics)

  def __add__(self, metrics: Iterable[MetricInformation]) -> 'MetricsConfig':
    return MetricsConfig(self._metrics + list(metrics))

  def of_type(self, include: MetricType) -> 'MetricsConfig':
    return MetricsConfig(m for m in self._metrics if m.type == include)

  def exclude_type(self, exclude: MetricType) -> 'MetricsConfig':
    return MetricsConfig(m for m in self._metrics if m.type != exclude)

  def append(self, metric: MetricInformation):
    self._metrics.append(metric)
    self._assert_unique()

  def extend(self, metrics: Iterable[MetricInformation]):
    for m in metrics:
      self.append(m)

  @property
  def is_single_objective(self) -> bool:
    return len(self.of_type(MetricType.OBJECTIVE)) == 1

  @property
  def is_safety_metric(self) -> bool:
    return bool(len(self.of_type(MetricType.SAFETY)))

  def __eq__(self, other) -> bool:
    if isinstance(other, MetricsConfig):
      return self._metrics == other._metrics
    return NotImplemented

  def __repr__(self) -> str:
    return f'MetricsConfig({self._metrics!r})'


class ProblemStatement:
  """Search space + metrics + metadata: the algorithm-facing study config."""

  def __init__(self,
               search_space: Optional[SearchSpace] = None,
               metric_information: Optional[
                   Iterable[MetricInformation]] = None,
               metadata: Optional[Metadata] = None):
    self.search_space = search_space if search_space is not None \
        else SearchSpace()
    if isinstance(metric_information, MetricsConfig):
      self.metric_information = metric_information
    else:
      self.metric_information = MetricsConfig(metric_information or ())
    self.metadata = metadata if metadata is not None else Metadata()

  @property
  def debug_info(self) -> str:
    return (f'ProblemStatement with {len(self.search_space.parameters)} '
            f'parameters and metrics {self.metric_information}')

  @classmethod
  def from_problem(cls: Type[_T], problem: 'ProblemStatement') -> _T:
    out = cls.__new__(cls)
    ProblemStatement.__init__(out, problem.search_space,
                              problem.metric_information, problem.metadata)
    return out

  def to_problem(self) -> 'ProblemStatement':
    return ProblemStatement(self.search_space, self.metric_information,
                            self.metadata)

  @property
  def is_single_objective(self) -> bool:
    return self.metric_information.is_single_objective

  @property
  def single_objective_metric_name(self) -> Optional[str]:
    objs = list(self.metric_information.of_type(MetricType.OBJECTIVE))
    if len(objs) == 1:
      return objs[0].name
    return None

  @property
  def is_safety_metric(self) -> bool:
    return self.metric_information.is_safety_metric

  def __eq__(self, other) -> bool:
    if not isinstance(other, ProblemStatement):
      return NotImplemented
    return (self.search_space == other.search_space and
            self.metric_information == other.metric_information and
            self.metadata == other.metadata)

  def __repr__(self) -> str:
    return (f'ProblemStatement(search_space={self.search_space!r}, '
            f'metric_information={self.metric_information!r})')
