"""Singleton-parameter stripping policy wrapper.

Capability parity with vizier/_src/pythia/singleton_params.py: search
spaces sometimes contain parameters with exactly one feasible value;
this wrapper removes them before the inner policy sees the problem and
re-attaches the constant values to every suggestion.
"""

from __future__ import annotations

import copy
from typing import Callable, Dict

from vizier_amd import pyvizier as vz
from vizier_amd._src.pythia.policy import (
    Policy,
    StudyDescriptor,
    SuggestDecision,
    SuggestRequest,
)


def _singleton_value(pc: vz.ParameterConfig):
  if pc.type == vz.ParameterType.DOUBLE:
    lo, hi = pc.bounds
    return lo if lo == hi else None
  values = pc.feasible_values
  return values[0] if len(values) == 1 else None


class SingletonParameterPolicyWrapper(Policy):
  """Strips single-valued parameters around an inner policy."""

  def __init__(self, policy_factory: Callable[..., Policy], *args,
               **kwargs):
    self._factory = policy_factory
    self._args = args
    self._kwargs = kwargs

  def suggest(self, request: SuggestRequest) -> SuggestDecision:
    config = request.study_config
    singletons: Dict[str, vz.ParameterValueTypes] = {}
    stripped = vz.SearchSpace()
    for pc in config.search_space.parameters:
      value = _singleton_value(pc)
      if value is not None:
        singletons[pc.name] = value
      else:
        stripped.add(pc)
    if not singletons:
      return self._factory(*self._args, **self._kwargs).suggest(request)

    inner_config = copy.deepcopy(config)
    inner_config.search_space = stripped
    inner_request = SuggestRequest(
        study_descriptor=StudyDescriptor(
            config=inner_config, guid=request.study_guid,
            max_trial_id=request.max_trial_id),
        count=request.count, checkpoint_dir=request.checkpoint_dir)
    decision = self._factory(*self._args, **self._kwargs).suggest(
        inner_request)
    for s in decision.suggestions:
      for name, value in singletons.items():
        s.parameters[name] = value
    return decision
