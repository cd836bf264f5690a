"""Validator helpers for attrs/dataclass-style fields.

Capability parity with vizier/utils/attrs_utils.py (:27-107): small
callables usable as `attr.field(validator=...)` OR called directly.
They work with any (instance, attribute, value) convention, so they
also serve plain-dataclass __post_init__ checks.
"""

from __future__ import annotations

import re
from typing import Any, Callable, Collection, Optional


def assert_not_empty(instance: Any, attribute: Any, value: Any) -> None:
  del instance
  if not value:
    raise ValueError(f'{getattr(attribute, "name", attribute)} '
                     f'must not be empty; got {value!r}')


def assert_not_negative(instance: Any, attribute: Any,
                        value: Any) -> None:
  del instance
  if value < 0:
    raise ValueError(f'{getattr(attribute, "name", attribute)} '
                     f'must be non-negative; got {value!r}')


def assert_not_none(instance: Any, attribute: Any, value: Any) -> None:
  del instance
  if value is None:
    raise ValueError(f'{getattr(attribute, "name", attribute)} '
                     'must not be None')


def assert_between(low: float, high: float
                   ) -> Callable[[Any, Any, Any], None]:
  """Returns a validator asserting low <= value <= high."""

  def validator(instance: Any, attribute: Any, value: Any) -> None:
    del instance
    if not low <= value <= high:
      raise ValueError(f'{getattr(attribute, "name", attribute)} must '
                       f'be in [{low}, {high}]; got {value!r}')

  return validator


def assert_re_fullmatch(pattern: str
                        ) -> Callable[[Any, Any, Any], None]:
  """Returns a validator asserting the value fullmatches `pattern`."""
  compiled = re.compile(pattern)

  def validator(instance: Any, attribute: Any, value: Any) -> None:
    del instance
    if not compiled.fullmatch(value):
      raise ValueError(f'{getattr(attribute, "name", attribute)} must '
                       f'fullmatch {pattern!r}; got {value!r}')

  return validator


def shape_equals(instance_to_shape: Callable[
    [Any], Collection[Optional[int]]]):
  """Returns a validator asserting an array field's shape.

  `instance_to_shape(instance)` gives the expected shape; None entries
  match any extent (attrs_utils.py:70).
  """

  def validator(instance: Any, attribute: Any, value: Any) -> None:
    expected = tuple(instance_to_shape(instance))
    actual = tuple(value.shape)
    ok = len(actual) == len(expected) and all(
        e is None or a == e for a, e in zip(actual, expected))
    if not ok:
      raise ValueError(f'{getattr(attribute, "name", attribute)} shape '
                       f'{actual} does not match expected {expected}')

  return validator
