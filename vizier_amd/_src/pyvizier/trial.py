"""Trial lifecycle data model.

MI355X-native rewrite with the capabilities of
vizier/_src/pyvizier/shared/trial.py (TrialStatus :81, Metric :91,
ParameterValue :128, Measurement :276, ParameterDict :345,
TrialSuggestion :404, Trial :439, TrialFilter :638, MetadataDelta :685).
"""

from __future__ import annotations

import copy
import datetime
import enum
import math
from collections import abc
from typing import Any, Dict, Iterable, Iterator, List, Mapping, Optional, Union

from vizier_amd._src.pyvizier.common import Metadata, Namespace
from vizier_amd._src.pyvizier.parameter_config import (
    ExternalType,
    ParameterType,
)

ParameterValueTypes = Union[str, int, float, bool]

TRUE_VALUE = 'true'
FALSE_VALUE = 'false'


class TrialStatus(enum.Enum):
  """Lifecycle state of a Trial."""

  UNKNOWN = 'UNKNOWN'
  REQUESTED = 'REQUESTED'
  ACTIVE = 'ACTIVE'
  COMPLETED = 'COMPLETED'
  STOPPING = 'STOPPING'


class Metric:
  """A single named metric value with an optional stddev."""

  __slots__ = ('_value', '_std')

  def __init__(self, value: float, *, std: Optional[float] = None):
    self._value = float(value)
    if std is not None:
      std = float(std)
      if not std >= 0:
        raise ValueError('Metric std must be non-negative.')
    self._std = std

  @property
  def value(self) -> float:
    return self._value

  @property
  def std(self) -> Optional[float]:
    return self._std

  def __eq__(self, other) -> bool:
    if isinstance(other, Metric):
      return (self._value == other._value or
              (math.isnan(self._value) and math.isnan(other._value))) and \
          self._std == other._std
    return NotImplemented

  def __hash__(self):
    return hash((self._value, self._std))

  def __repr__(self) -> str:
    return (f'Metric(value={self._value}' +
            (f', std={self._std})' if self._std is not None else ')'))


NaNMetric = Metric(value=float('nan'))


class ParameterValue:
  """Immutable wrapper around a parameter value with casting accessors.

  The strings 'true'/'false' are treated as booleans when cast, per the
  reference behavior (vizier/_src/pyvizier/shared/trial.py:128).
  """

  __slots__ = ('_value',)

  def __init__(self, value: ParameterValueTypes):
    if not isinstance(value, (str, int, float, bool)):
      raise TypeError(f'Invalid parameter value {value!r}')
    object.__setattr__(self, '_value', value)

  def __setattr__(self, *_):
    raise AttributeError('ParameterValue is immutable')

  def __reduce__(self):
    # Supports copy/deepcopy/pickle despite the immutability guard.
    return (ParameterValue, (self._value,))

  @property
  def value(self) -> ParameterValueTypes:
    return self._value

  @property
  def as_float(self) -> Optional[float]:
    if isinstance(self._value, bool):
      return 1.0 if self._value else 0.0
    if isinstance(self._value, (int, float)):
      return float(self._value)
    if self._value == TRUE_VALUE:
      return 1.0
    if self._value == FALSE_VALUE:
      return 0.0
    try:
      return float(self._value)
    except ValueError:
      return None

  @property
  def as_int(self) -> Optional[int]:
    f = self.as_float
    if f is None or not float(f).is_integer():
      return None
    return int(f)

  @property
  def as_bool(self) -> Optional[bool]:
    if isinstance(self._value, bool):
      return self._value
    if isinstance(self._value, str):
      if self._value == TRUE_VALUE:
        return True
      if self._value == FALSE_VALUE:
        return False
      return None
    if isinstance(self._value, (int, float)):
      if self._value == 1:
        return True
      if self._value == 0:
        return False
    return None

  @property
  def as_str(self) -> Optional[str]:
    if isinstance(self._value, bool):
      return TRUE_VALUE if self._value else FALSE_VALUE
    if isinstance(self._value, str):
      return self._value
    return None

  def cast_as_internal(self, internal_type: ParameterType
                       ) -> ParameterValueTypes:
    internal_type.assert_correct_type(self._value)
    if internal_type in (ParameterType.DOUBLE, ParameterType.DISCRETE):
      return self.as_float
    if internal_type == ParameterType.INTEGER:
      return self.as_int
    if internal_type == ParameterType.CATEGORICAL:
      return self.as_str
    raise RuntimeError(f'Unknown type {internal_type}')

  def cast(self, external_type: ExternalType) -> ParameterValueTypes:
    if external_type == ExternalType.INTERNAL:
      return self._value
    if external_type == ExternalType.BOOLEAN:
      out = self.as_bool
    elif external_type == ExternalType.INTEGER:
      out = self.as_int
    elif external_type == ExternalType.FLOAT:
      out = self.as_float
    else:
      raise ValueError(f'Unknown external type {external_type}')
    if out is None:
      raise ValueError(f'Cannot cast {self._value!r} to {external_type}')
    return out

  def __eq__(self, other) -> bool:
    if isinstance(other, ParameterValue):
      return self._value == other._value
    return NotImplemented

  def __hash__(self):
    return hash(self._value)

  def __repr__(self) -> str:
    return f'ParameterValue({self._value!r})'


class ParameterDict(abc.MutableMapping):
  """Maps parameter names to ParameterValues; accepts raw values on assign."""

  def __init__(self, *args, **kwargs):
    self._items: Dict[str, ParameterValue] = {}
    for k, v in dict(*args, **kwargs).items():
      self[k] = v

  def __setitem__(self, key: str, value):
    if not isinstance(value, ParameterValue):
      value = ParameterValue(value)
    self._items[key] = value

  def __getitem__(self, key: str) -> ParameterValue:
    return self._items[key]

  def __delitem__(self, key: str):
    del self._items[key]

  def __iter__(self) -> Iterator[str]:
    return iter(self._items)

  def __len__(self) -> int:
    return len(self._items)

  def get_value(self, key: str, default=None) -> Optional[ParameterValueTypes]:
    pv = self._items.get(key)
    return pv.value if pv is not None else default

  def as_dict(self) -> Dict[str, ParameterValueTypes]:
    return {k: v.value for k, v in self._items.items()}

  def __eq__(self, other) -> bool:
    if isinstance(other, ParameterDict):
      return self._items == other._items
    if isinstance(other, dict):
      return self.as_dict() == other or self._items == other
    return NotImplemented

  def __repr__(self) -> str:
    return f'ParameterDict({self.as_dict()!r})'


class _MetricDict(dict):
  """dict[str, Metric] that accepts floats on assignment."""

  def __init__(self, *args, **kwargs):
    super().__init__()
    for k, v in dict(*args, **kwargs).items():
      self[k] = v

  def __setitem__(self, key: str, value):
    if not isinstance(value, Metric):
      value = Metric(value=float(value))
    super().__setitem__(key, value)


class Measurement:
  """A set of metric values observed at a point in a trial's evaluation."""

  def __init__(self, metrics: Optional[Mapping[str, Union[float, Metric]]]
               = None, *, elapsed_secs: float = 0, steps: int = 0,
               checkpoint_path: str = ''):
    self._metrics = _MetricDict(metrics or {})
    elapsed_secs = float(elapsed_secs)
    if not (math.isfinite(elapsed_secs) and elapsed_secs >= 0):
      raise ValueError('elapsed_secs must be finite and non-negative.')
    self.elapsed_secs = elapsed_secs
    steps = int(steps)
    if steps < 0:
      raise ValueError('steps must be non-negative.')
    self.steps = steps
    self.checkpoint_path = checkpoint_path

  @property
  def metrics(self) -> Dict[str, Metric]:
    return self._metrics

  @metrics.setter
  def metrics(self, value: Mapping[str, Union[float, Metric]]):
    self._metrics = _MetricDict(value)

  def as_float_dict(self) -> Dict[str, float]:
    return {k: m.value for k, m in self._metrics.items()}

  def __eq__(self, other) -> bool:
    if not isinstance(other, Measurement):
      return NotImplemented
    return (self._metrics == other._metrics and
            self.elapsed_secs == other.elapsed_secs and
            self.steps == other.steps and
            self.checkpoint_path == other.checkpoint_path)

  def __repr__(self) -> str:
    return (f'Measurement(metrics={dict(self._metrics)!r}, '
            f'elapsed_secs={self.elapsed_secs}, steps={self.steps})')


def _localize(dt: Optional[datetime.datetime]) -> Optional[datetime.datetime]:
  return dt.astimezone() if dt else None


class TrialSuggestion:
  """A suggested set of parameters, not yet assigned a trial id."""

  def __init__(self, parameters: Optional[Mapping[str, Any]] = None, *,
               metadata: Optional[Metadata] = None):
    if isinstance(parameters, ParameterDict):
      self.parameters = parameters
    else:
      self.parameters = ParameterDict(parameters or {})
    self.metadata = metadata if metadata is not None else Metadata()

  def to_trial(self, uid: int = 0) -> 'Trial':
    return Trial(id=uid, parameters=self.parameters, metadata=self.metadata)

  def __eq__(self, other) -> bool:
    if type(other) is not type(self):
      return NotImplemented
    return (self.parameters == other.parameters and
            self.metadata == other.metadata)

  def __repr__(self) -> str:
    return f'TrialSuggestion(parameters={self.parameters!r})'


class Trial(TrialSuggestion):
  """A parameter assignment plus its evaluation state."""

  def __init__(self, parameters: Optional[Mapping[str, Any]] = None, *,
               id: int = 0,  # pylint: disable=redefined-builtin
               metadata: Optional[Metadata] = None,
               is_requested: bool = False,
               assigned_worker: Optional[str] = None,
               stopping_reason: Optional[str] = None,
               infeasibility_reason: Optional[str] = None,
               description: Optional[str] = None,
               related_links: Optional[Dict[str, str]] = None,
               final_measurement: Optional[Measurement] = None,
               measurements: Optional[List[Measurement]] = None,
               creation_time: Optional[datetime.datetime] = None,
               completion_time: Optional[datetime.datetime] = None):
    super().__init__(parameters, metadata=metadata)
    self.id = int(id)
    self.is_requested = bool(is_requested)
    self.assigned_worker = assigned_worker
    self.stopping_reason = stopping_reason
    self._infeasibility_reason = infeasibility_reason
    self.description = description
    self.related_links = dict(related_links or {})
    self.final_measurement = final_measurement
    self.measurements = list(measurements or [])
    self.creation_time = _localize(creation_time or datetime.datetime.now())
    self.completion_time = _localize(completion_time)
    if self.completion_time is None and (self.final_measurement is not None or
                                         self.infeasible):
      self.completion_time = self.creation_time

  @property
  def duration(self) -> Optional[datetime.timedelta]:
    if self.completion_time:
      return self.completion_time - self.creation_time
    return None

  @property
  def status(self) -> TrialStatus:
    if self.final_measurement is not None or self.infeasible:
      return TrialStatus.COMPLETED
    if self.stopping_reason is not None:
      return TrialStatus.STOPPING
    if self.is_requested:
      return TrialStatus.REQUESTED
    return TrialStatus.ACTIVE

  @property
  def is_completed(self) -> bool:
    return (self.status == TrialStatus.COMPLETED or
            self.completion_time is not None)

  @property
  def infeasible(self) -> bool:
    return self._infeasibility_reason is not None

  @property
  def infeasibility_reason(self) -> Optional[str]:
    return self._infeasibility_reason

  @property
  def final_measurement_or_die(self) -> Measurement:
    if self.final_measurement is None:
      raise ValueError(f'Trial is missing final_measurement: {self}')
    return self.final_measurement

  def complete(self, measurement: Measurement, *,
               infeasibility_reason: Optional[str] = None,
               inplace: bool = True) -> 'Trial':
    if not inplace:
      return copy.deepcopy(self).complete(
          measurement, infeasibility_reason=infeasibility_reason, inplace=True)
    self.final_measurement = copy.deepcopy(measurement)
    if infeasibility_reason is not None:
      self._infeasibility_reason = infeasibility_reason
    self.completion_time = _localize(datetime.datetime.now())
    return self

  def __eq__(self, other) -> bool:
    if not isinstance(other, Trial):
      return NotImplemented
    return (self.id == other.id and self.parameters == other.parameters and
            self.metadata == other.metadata and
            self.final_measurement == other.final_measurement and
            self.measurements == other.measurements and
            self._infeasibility_reason == other._infeasibility_reason and
            self.is_requested == other.is_requested and
            self.stopping_reason == other.stopping_reason and
            self.creation_time == other.creation_time)

  def __repr__(self) -> str:
    return (f'Trial(id={self.id}, status={self.status.value}, '
            f'parameters={self.parameters.as_dict()!r})')


# Aliases kept for API parity with the reference.
CompletedTrial = Trial
PendingTrial = Trial
CompletedTrialWithMeasurements = Trial
PendingTrialWithMeasurements = Trial


class TrialFilter:
  """Predicate over trials (ids / min_id / max_id / status)."""

  def __init__(self, *, ids: Optional[Iterable[int]] = None,
               min_id: Optional[int] = None, max_id: Optional[int] = None,
               status: Optional[Iterable[TrialStatus]] = None):
    self.ids = frozenset(ids) if ids is not None else None
    self.min_id = min_id
    self.max_id = max_id
    if isinstance(status, (TrialStatus, str)):
      status = [status]  # a lone status is a common call-site shape
    self.status = frozenset(TrialStatus(s) if not isinstance(s, TrialStatus)
                            else s for s in status) if status else None

  def __call__(self, trial: Trial) -> bool:
    if self.ids is not None and trial.id not in self.ids:
      return False
    if self.min_id is not None and trial.id < self.min_id:
      return False
    if self.max_id is not None and trial.id > self.max_id:
      return False
    if self.status is not None and trial.status not in self.status:
      return False
    return True


class MetadataDelta:
  """Metadata changes to apply to a study and/or its trials.

  `on_study` carries study-level metadata; `on_trials[trial_id]` carries
  per-trial metadata. Mirrors vizier/_src/pyvizier/shared/trial.py:685.
  """

  def __init__(self, on_study: Optional[Metadata] = None,
               on_trials: Optional[Dict[int, Metadata]] = None):
    self.on_study = on_study if on_study is not None else Metadata()
    self.on_trials: Dict[int, Metadata] = dict(on_trials or {})
    # defaultdict-like behavior for convenience
  def assign(self, namespace: str, key: str, value,
             *, trial: Optional[Trial] = None, trial_id: Optional[int] = None):
    """Assigns metadata either on the study or on a trial."""
    if trial is not None and trial_id is not None and trial.id != trial_id:
      raise ValueError('Both trial and trial_id given but they disagree.')
    if trial is not None:
      trial_id = trial.id
    if trial_id is None:
      self.on_study.abs_ns(Namespace.decode(namespace))[key] = value
    else:
      md = self.on_trials.setdefault(trial_id, Metadata())
      md.abs_ns(Namespace.decode(namespace))[key] = value

  def __bool__(self) -> bool:
    return bool(self.on_study) or any(bool(m) for m in self.on_trials.values())
