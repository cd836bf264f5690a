"""Parameter configurations and search spaces.

MI355X-native rewrite with the capabilities of
vizier/_src/pyvizier/shared/parameter_config.py (ParameterConfig :168,
SearchSpaceSelector :794, SearchSpace :1298). Same public surface
(`add_float_param`, conditional `select(...)`, value casting), new
implementation.
"""

from __future__ import annotations

import copy
import enum
import math
import re
from typing import Any, Dict, Iterable, Iterator, List, Optional, Sequence, Tuple, Union

ParameterValueTypes = Union[str, int, float, bool]
MonotypeParameterSequence = Sequence[ParameterValueTypes]

_EPSILON = 1e-10


class ParameterType(enum.Enum):
  """The type of a parameter."""

  DOUBLE = 'DOUBLE'
  INTEGER = 'INTEGER'
  CATEGORICAL = 'CATEGORICAL'
  DISCRETE = 'DISCRETE'
  CUSTOM = 'CUSTOM'

  def is_numeric(self) -> bool:
    return self in (ParameterType.DOUBLE, ParameterType.INTEGER,
                    ParameterType.DISCRETE)

  def is_continuous(self) -> bool:
    return self == ParameterType.DOUBLE

  def assert_correct_type(self, value: ParameterValueTypes) -> None:
    if self.is_numeric():
      if not isinstance(value, (int, float)) or isinstance(value, bool):
        raise TypeError(f'Type {self} expects a number; got {value!r}')
      if self == ParameterType.INTEGER and float(value) != int(value):
        raise TypeError(f'Type {self} expects an integral value; got {value!r}')
    elif self == ParameterType.CATEGORICAL:
      if not isinstance(value, str):
        raise TypeError(f'Type {self} expects str; got {value!r}')


class ScaleType(enum.Enum):
  """How a numeric parameter is scaled into the model's [0, 1] range."""

  LINEAR = 'LINEAR'
  LOG = 'LOG'
  REVERSE_LOG = 'REVERSE_LOG'
  UNIFORM_DISCRETE = 'UNIFORM_DISCRETE'

  def is_nonlinear(self) -> bool:
    return self in (ScaleType.LOG, ScaleType.REVERSE_LOG)


class ExternalType(enum.Enum):
  """The type the client presents to callers (not used by the service)."""

  INTERNAL = 'INTERNAL'
  BOOLEAN = 'BOOLEAN'
  INTEGER = 'INTEGER'
  FLOAT = 'FLOAT'


class FidelityMode(enum.Enum):
  """Experimental fidelity semantics (kept for API parity)."""

  SEQUENTIAL_CONDITIONAL = 'SEQUENTIAL_CONDITIONAL'
  NESTED = 'NESTED'


class FidelityConfig:
  """Experimental fidelity configuration (API parity only)."""

  def __init__(self, mode: FidelityMode, relative_cost: Sequence[float] = ()):
    self.mode = FidelityMode(mode)
    self.relative_cost = tuple(relative_cost)


class InvalidParameterError(ValueError):
  """Raised when a parameter value is not feasible for its config."""


def _is_integral(v: float) -> bool:
  return math.isclose(v, round(v), abs_tol=_EPSILON)


class ParameterConfig:
  """Immutable-ish configuration of a single parameter (plus conditionals).

  Construct with `ParameterConfig.factory(...)`:
    * bounds=(lo, hi) with float endpoints -> DOUBLE
    * bounds=(lo, hi) with int endpoints   -> INTEGER
    * feasible_values=[str, ...]           -> CATEGORICAL
    * feasible_values=[numbers...]         -> DISCRETE
  """

  def __init__(self, *, name: str, type: ParameterType,
               bounds: Optional[Tuple[float, float]] = None,
               feasible_values: Optional[Tuple[ParameterValueTypes, ...]] = None,
               scale_type: Optional[ScaleType] = None,
               default_value: Optional[ParameterValueTypes] = None,
               external_type: ExternalType = ExternalType.INTERNAL,
               fidelity_config: Optional[FidelityConfig] = None):
    if not name:
      raise ValueError('Parameter name cannot be empty.')
    self._name = name
    self._type = type
    self._bounds = bounds
    self._feasible_values = tuple(feasible_values) if feasible_values else None
    self._scale_type = scale_type
    self._default_value = default_value
    self._external_type = external_type or ExternalType.INTERNAL
    self.fidelity_config = fidelity_config
    # parent value -> subspace of child parameters.
    self._subspaces: Dict[ParameterValueTypes, 'SearchSpace'] = {}

  # -- construction --------------------------------------------------------

  @classmethod
  def factory(
      cls,
      name: str,
      *,
      bounds: Optional[Tuple[Union[int, float], Union[int, float]]] = None,
      feasible_values: Optional[MonotypeParameterSequence] = None,
      children: Optional[Sequence[Tuple[MonotypeParameterSequence,
                                        'ParameterConfig']]] = None,
      scale_type: Optional[ScaleType] = None,
      default_value: Optional[ParameterValueTypes] = None,
      external_type: ExternalType = ExternalType.INTERNAL,
      fidelity_config: Optional[FidelityConfig] = None,
  ) -> 'ParameterConfig':
    """Creates a ParameterConfig; infers the type from bounds/feasible_values."""
    if (bounds is None) == (feasible_values is None):
      raise ValueError(
          f'Exactly one of bounds/feasible_values must be given for {name}: '
          f'bounds={bounds} feasible_values={feasible_values}')
    if bounds is not None:
      lo, hi = bounds
      if lo > hi:
        raise ValueError(f'min {lo} > max {hi} for parameter {name}')
      if isinstance(lo, bool) or isinstance(hi, bool):
        raise ValueError(f'Bool bounds are invalid for {name}')
      if isinstance(lo, int) and isinstance(hi, int):
        ptype = ParameterType.INTEGER
        bounds = (int(lo), int(hi))
      else:
        ptype = ParameterType.DOUBLE
        bounds = (float(lo), float(hi))
      feasible = None
    else:
      feasible_values = tuple(feasible_values)
      if len(set(feasible_values)) != len(feasible_values):
        raise ValueError(f'Duplicate feasible values for {name}: '
                         f'{feasible_values}')
      if all(isinstance(v, str) for v in feasible_values):
        ptype = ParameterType.CATEGORICAL
        feasible = tuple(sorted(feasible_values))
      elif all(isinstance(v, (int, float)) and not isinstance(v, bool)
               for v in feasible_values):
        ptype = ParameterType.DISCRETE
        feasible = tuple(sorted(float(v) for v in feasible_values))
      else:
        raise ValueError(
            f'Feasible values must be all-numeric or all-str for {name}: '
            f'{feasible_values}')
      bounds = None

    pc = cls(name=name, type=ptype, bounds=bounds, feasible_values=feasible,
             scale_type=scale_type, default_value=None,
             external_type=external_type, fidelity_config=fidelity_config)
    if default_value is not None:
      pc._default_value = pc.cast_value(default_value)
      if not pc.contains(pc._default_value):
        raise InvalidParameterError(
            f'Default value {default_value} is infeasible for {name}')
    if children:
      for parent_values, child in children:
        for v in parent_values:
          pc.add_child(v, child)
    return pc

  def clone_without_children(self) -> 'ParameterConfig':
    pc = ParameterConfig(
        name=self._name, type=self._type, bounds=self._bounds,
        feasible_values=self._feasible_values, scale_type=self._scale_type,
        default_value=self._default_value, external_type=self._external_type,
        fidelity_config=self.fidelity_config)
    return pc

  # -- basic accessors -----------------------------------------------------

  @property
  def name(self) -> str:
    return self._name

  @property
  def type(self) -> ParameterType:
    return self._type

  @property
  def scale_type(self) -> Optional[ScaleType]:
    return self._scale_type

  @property
  def external_type(self) -> ExternalType:
    return self._external_type

  @property
  def default_value(self) -> Optional[ParameterValueTypes]:
    return self._default_value

  @property
  def bounds(self) -> Tuple[float, float]:
    """(min, max) for numeric types (DISCRETE bounds from feasible values)."""
    if self._bounds is not None:
      return self._bounds
    if self._type == ParameterType.DISCRETE:
      return (self._feasible_values[0], self._feasible_values[-1])
    raise ValueError(f'Parameter {self._name} of type {self._type} '
                     'has no bounds.')

  @property
  def feasible_values(self) -> List[ParameterValueTypes]:
    if self._feasible_values is not None:
      return list(self._feasible_values)
    if self._type == ParameterType.INTEGER:
      lo, hi = self._bounds
      return list(range(int(lo), int(hi) + 1))
    raise ValueError(f'Parameter {self._name} of type {self._type} '
                     'has no finite feasible values.')

  @property
  def num_feasible_values(self) -> Union[int, float]:
    if self._type == ParameterType.DOUBLE:
      return float('inf')
    if self._type == ParameterType.INTEGER:
      lo, hi = self._bounds
      return int(hi) - int(lo) + 1
    return len(self._feasible_values)

  # -- conditional children ------------------------------------------------

  def add_child(self, parent_value: ParameterValueTypes,
                child: 'ParameterConfig') -> 'ParameterConfig':
    """Adds `child` as a conditional parameter active when value==parent_value."""
    parent_value = self.cast_value(parent_value)
    if not self.contains(parent_value):
      raise InvalidParameterError(
          f'{parent_value!r} is not feasible for {self._name}; cannot attach '
          f'child {child.name}')
    sub = self._subspaces.setdefault(parent_value, SearchSpace())
    return sub.add(copy.deepcopy(child))

  def subspace(self, parent_value: ParameterValueTypes) -> 'SearchSpace':
    parent_value = self.cast_value(parent_value)
    if not self.contains(parent_value):
      raise InvalidParameterError(
          f'{parent_value!r} is not feasible for {self._name}')
    return self._subspaces.setdefault(parent_value, SearchSpace())

  @property
  def subspaces_by_value(self) -> Dict[ParameterValueTypes, 'SearchSpace']:
    return dict(self._subspaces)

  @property
  def child_parameter_configs(self) -> List['ParameterConfig']:
    out = []
    for sub in self._subspaces.values():
      out.extend(sub.parameters)
    return out

  def traverse(self, show_children: bool = True) -> Iterator['ParameterConfig']:
    """Yields this config and (recursively) all conditional children."""
    yield self
    for sub in self._subspaces.values():
      for p in sub.parameters:
        yield from p.traverse(show_children)

  # -- value handling ------------------------------------------------------

  def cast_value(self, value: ParameterValueTypes) -> ParameterValueTypes:
    """Casts `value` to this parameter's canonical Python type."""
    if self._type == ParameterType.DOUBLE:
      return float(value)
    if self._type == ParameterType.INTEGER:
      f = float(value)
      if not _is_integral(f):
        raise InvalidParameterError(
            f'{value!r} is not integral for INTEGER parameter {self._name}')
      return int(round(f))
    if self._type == ParameterType.DISCRETE:
      return float(value)
    if self._type == ParameterType.CATEGORICAL:
      if isinstance(value, bool):
        return 'true' if value else 'false'
      return str(value)
    return value

  def contains(self, value: ParameterValueTypes) -> bool:
    try:
      value = self.cast_value(value)
    except (InvalidParameterError, ValueError, TypeError):
      return False
    if self._type == ParameterType.DOUBLE:
      lo, hi = self._bounds
      return lo <= value <= hi
    if self._type == ParameterType.INTEGER:
      lo, hi = self._bounds
      return lo <= value <= hi
    if self._type == ParameterType.DISCRETE:
      return any(abs(value - v) <= _EPSILON for v in self._feasible_values)
    if self._type == ParameterType.CATEGORICAL:
      return value in self._feasible_values
    return True

  def round_to_feasible(self, value: float) -> ParameterValueTypes:
    """Rounds a continuous value to the nearest feasible point."""
    if self._type == ParameterType.DOUBLE:
      lo, hi = self._bounds
      return min(max(float(value), lo), hi)
    if self._type == ParameterType.INTEGER:
      lo, hi = self._bounds
      return int(min(max(round(value), lo), hi))
    if self._type == ParameterType.DISCRETE:
      return min(self._feasible_values, key=lambda v: abs(v - value))
    raise ValueError(f'Cannot round {self._type} parameter {self._name}')

  # -- merging (for ParameterConfigSelector.merge) -------------------------

  @classmethod
  def merge(cls, a: 'ParameterConfig', b: 'ParameterConfig') -> 'ParameterConfig':
    """Union of the feasible sets of two configs with the same name/type."""
    if a.name != b.name or a.type != b.type:
      raise ValueError(f'Cannot merge {a.name}({a.type}) with '
                       f'{b.name}({b.type})')
    if a._bounds is not None:
      bounds = (min(a._bounds[0], b._bounds[0]), max(a._bounds[1],
                                                     b._bounds[1]))
      if a.type == ParameterType.INTEGER:
        bounds = (int(bounds[0]), int(bounds[1]))
      return cls.factory(a.name, bounds=bounds, scale_type=a.scale_type,
                         external_type=a.external_type)
    feas = tuple(sorted(set(a._feasible_values) | set(b._feasible_values)))
    return cls.factory(a.name, feasible_values=feas, scale_type=a.scale_type,
                       external_type=a.external_type)

  def __eq__(self, other) -> bool:
    if not isinstance(other, ParameterConfig):
      return NotImplemented
    return (self._name == other._name and self._type == other._type and
            self._bounds == other._bounds and
            self._feasible_values == other._feasible_values and
            self._scale_type == other._scale_type and
            self._default_value == other._default_value and
            self._subspaces == other._subspaces)

  def __repr__(self) -> str:
    dom = (f'bounds={self._bounds}' if self._bounds is not None
           else f'feasible={self._feasible_values}')
    return (f'ParameterConfig(name={self._name!r}, type={self._type.value}, '
            f'{dom}, scale={self._scale_type}, default={self._default_value})')


class ParameterConfigSelector:
  """A selection of ParameterConfigs (supports conditional-space building)."""

  def __init__(self, selected: Union[ParameterConfig,
                                     Iterable[ParameterConfig]]):
    if isinstance(selected, ParameterConfig):
      self._selected: Tuple[ParameterConfig, ...] = (selected,)
    else:
      self._selected = tuple(selected)

  def __iter__(self) -> Iterator[ParameterConfig]:
    return iter(self._selected)

  def __len__(self) -> int:
    return len(self._selected)

  def select_values(self, values: MonotypeParameterSequence
                    ) -> 'SearchSpaceSelector':
    """Selects the subspaces under the given parent values."""
    values = tuple(values)
    for v in values:
      for cfg in self._selected:
        if not cfg.contains(v):
          raise InvalidParameterError(f'{v!r} is not feasible in {cfg}')
    spaces = []
    for v in values:
      for cfg in self._selected:
        spaces.append(cfg.subspace(v))
    return SearchSpaceSelector(spaces)

  def merge(self) -> 'ParameterConfigSelector':
    merged: Dict[str, ParameterConfig] = {}
    for cfg in self._selected:
      if cfg.name in merged:
        merged[cfg.name] = ParameterConfig.merge(merged[cfg.name], cfg)
      else:
        merged[cfg.name] = cfg
    return ParameterConfigSelector(merged.values())


_MULTIDIM_RE = re.compile(r'(?P<name>[^()]*)\[(?P<index>\d+)\]$')


class SearchSpaceSelector:
  """Holds references to one or more (sub)spaces and adds parameters to them."""

  def __init__(self, selected: Union['SearchSpace', Iterable['SearchSpace']]):
    if isinstance(selected, SearchSpace):
      self._selected: Tuple['SearchSpace', ...] = (selected,)
    else:
      self._selected = tuple(selected)

  def __len__(self) -> int:
    return len(self._selected)

  @staticmethod
  def parse_multi_dimensional_parameter_name(
      name: str) -> Optional[Tuple[str, int]]:
    m = _MULTIDIM_RE.match(name)
    if m is None:
      return None
    return (m.group('name'), int(m.group('index')))

  def _names(self, name: str, index: Optional[int]) -> List[str]:
    if not name:
      raise ValueError('Parameter name cannot be empty.')
    if index is None:
      return [name]
    if index < 0:
      raise ValueError(f'index must be >= 0; got {index}')
    return [f'{name}[{index}]']

  def _add(self, configs: Iterable[ParameterConfig]) -> ParameterConfigSelector:
    added = []
    for cfg in configs:
      for space in self._selected:
        added.append(space.add(copy.deepcopy(cfg)))
    return ParameterConfigSelector(added)

  def add_float_param(self, name: str, min_value: float, max_value: float, *,
                      default_value: Optional[float] = None,
                      scale_type: Optional[ScaleType] = ScaleType.LINEAR,
                      index: Optional[int] = None) -> ParameterConfigSelector:
    cfgs = [ParameterConfig.factory(
        n, bounds=(float(min_value), float(max_value)),
        default_value=default_value, scale_type=scale_type)
            for n in self._names(name, index)]
    return self._add(cfgs)

  def add_int_param(self, name: str, min_value: int, max_value: int, *,
                    default_value: Optional[int] = None,
                    scale_type: Optional[ScaleType] = None,
                    index: Optional[int] = None,
                    experimental_fidelity_config: Optional[FidelityConfig] = None
                    ) -> ParameterConfigSelector:
    if not _is_integral(float(min_value)):
      raise ValueError(f'min_value must be an integer; got {min_value}')
    if not _is_integral(float(max_value)):
      raise ValueError(f'max_value must be an integer; got {max_value}')
    cfgs = [ParameterConfig.factory(
        n, bounds=(int(min_value), int(max_value)),
        default_value=default_value, scale_type=scale_type,
        fidelity_config=experimental_fidelity_config)
            for n in self._names(name, index)]
    return self._add(cfgs)

  def add_discrete_param(self, name: str,
                         feasible_values: Sequence[Union[int, float]], *,
                         default_value: Optional[Union[int, float]] = None,
                         scale_type: Optional[ScaleType] = ScaleType.LINEAR,
                         index: Optional[int] = None,
                         auto_cast: bool = True) -> ParameterConfigSelector:
    if not auto_cast:
      ext = ExternalType.INTERNAL
    elif all(isinstance(v, int) or float(v).is_integer()
             for v in feasible_values):
      ext = ExternalType.INTEGER
    else:
      ext = ExternalType.FLOAT
    cfgs = [ParameterConfig.factory(
        n, feasible_values=[float(v) for v in feasible_values],
        default_value=default_value, scale_type=scale_type, external_type=ext)
            for n in self._names(name, index)]
    return self._add(cfgs)

  def add_categorical_param(self, name: str, feasible_values: Sequence[str], *,
                            default_value: Optional[str] = None,
                            scale_type: Optional[ScaleType] = None,
                            index: Optional[int] = None
                            ) -> ParameterConfigSelector:
    cfgs = [ParameterConfig.factory(
        n, feasible_values=list(feasible_values), default_value=default_value,
        scale_type=scale_type)
            for n in self._names(name, index)]
    return self._add(cfgs)

  def add_bool_param(self, name: str, feasible_values: Optional[
      Sequence[bool]] = None, *, default_value: Optional[bool] = None,
                     index: Optional[int] = None) -> ParameterConfigSelector:
    if feasible_values is None:
      feasible_values = (True, False)
    str_values = ['true' if v else 'false' for v in feasible_values]
    str_default = None
    if default_value is not None:
      str_default = 'true' if default_value else 'false'
    cfgs = [ParameterConfig.factory(
        n, feasible_values=str_values, default_value=str_default,
        external_type=ExternalType.BOOLEAN)
            for n in self._names(name, index)]
    return self._add(cfgs)

  def select(self, parameter_name: str,
             parameter_values: Optional[MonotypeParameterSequence] = None):
    """Selects a parameter (values=None) or conditional subspaces."""
    configs = [space.get(parameter_name) for space in self._selected]
    if parameter_values is None:
      return ParameterConfigSelector(configs)
    return ParameterConfigSelector(configs).select_values(parameter_values)

  def select_all(self) -> ParameterConfigSelector:
    out = []
    for space in self._selected:
      for cfg in space.parameters:
        out.extend(cfg.traverse())
    return ParameterConfigSelector(out)


class SearchSpace:
  """An ordered collection of ParameterConfigs, possibly conditional."""

  def __init__(self, parameters: Iterable[ParameterConfig] = ()):
    self._parameters: List[ParameterConfig] = []
    for p in parameters:
      self.add(p)

  @property
  def parameters(self) -> List[ParameterConfig]:
    return list(self._parameters)

  @property
  def parameter_names(self) -> List[str]:
    return [p.name for p in self._parameters]

  def add(self, config: ParameterConfig,
          *, replace: bool = False) -> ParameterConfig:
    existing = next((p for p in self._parameters if p.name == config.name),
                    None)
    if existing is not None:
      if not replace:
        raise ValueError(
            f'Duplicate parameter name {config.name!r} in search space.')
      self._parameters[self._parameters.index(existing)] = config
      return config
    self._parameters.append(config)
    return config

  def get(self, name: str) -> ParameterConfig:
    for p in self._parameters:
      if p.name == name:
        return p
    raise KeyError(f'No parameter named {name!r} in search space.')

  def pop(self, name: str) -> ParameterConfig:
    p = self.get(name)
    self._parameters.remove(p)
    return p

  def __contains__(self, name: str) -> bool:
    return any(p.name == name for p in self._parameters)

  def __len__(self) -> int:
    """Number of top-level parameters (reference-compatible)."""
    return len(self._parameters)

  def num_parameters(self,
                     param_type: Optional[ParameterType] = None) -> int:
    """Counts parameters of the given type across the whole tree."""
    count = 0
    for top in self._parameters:
      for cfg in top.traverse():
        if param_type is None or cfg.type == param_type:
          count += 1
    return count

  @property
  def is_conditional(self) -> bool:
    return any(p.child_parameter_configs for p in self._parameters)

  def select_root(self) -> SearchSpaceSelector:
    return SearchSpaceSelector(self)

  @property
  def root(self) -> SearchSpaceSelector:
    return SearchSpaceSelector(self)

  def __eq__(self, other) -> bool:
    if not isinstance(other, SearchSpace):
      return NotImplemented
    return self._parameters == other._parameters

  def __repr__(self) -> str:
    return f'SearchSpace({self._parameters!r})'
