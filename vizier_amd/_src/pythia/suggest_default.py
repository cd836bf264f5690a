"""Default/center-point seeding.

Capability parity with vizier/_src/pythia/suggest_default.py
(get_default_parameters :33, seed_with_default :57).
"""

from __future__ import annotations

import functools
import math
from typing import Dict

from vizier_amd._src.pyvizier.parameter_config import (
    ParameterConfig,
    ParameterType,
    ScaleType,
    SearchSpace,
)
from vizier_amd._src.pyvizier.trial import ParameterDict
from vizier_amd._src.pythia.policy import SuggestDecision, SuggestRequest


def _default_value(pc: ParameterConfig):
  """The parameter's default, or the center of its feasible set."""
  if pc.default_value is not None:
    return pc.default_value
  if pc.type == ParameterType.DOUBLE:
    lo, hi = pc.bounds
    if pc.scale_type == ScaleType.LOG and lo > 0:
      return float(math.exp(0.5 * (math.log(lo) + math.log(hi))))
    return (lo + hi) / 2.0
  if pc.type == ParameterType.INTEGER:
    lo, hi = pc.bounds
    return int((lo + hi) // 2)
  values = pc.feasible_values
  return values[(len(values) - 1) // 2]


def get_default_parameters(search_space: SearchSpace) -> ParameterDict:
  """Center/default assignment, descending into active conditionals."""
  params = ParameterDict()

  def visit(configs):
    for pc in configs:
      value = _default_value(pc)
      params[pc.name] = value
      sub = pc.subspaces_by_value.get(pc.cast_value(value))
      if sub is not None:
        visit(sub.parameters)

  visit(search_space.parameters)
  return params


def seed_with_default(policy_cls):
  """Class decorator: first suggestion == the search-space default."""
  original_suggest = policy_cls.suggest

  @functools.wraps(original_suggest)
  def suggest(self, request: SuggestRequest) -> SuggestDecision:
    if request.max_trial_id == 0 and request.count >= 1:
      from vizier_amd._src.pyvizier.trial import TrialSuggestion
      default = TrialSuggestion(
          get_default_parameters(request.study_config.search_space))
      rest = original_suggest(self, SuggestRequest(
          study_descriptor=request.study_descriptor,
          count=request.count - 1,
          checkpoint_dir=request.checkpoint_dir)) if request.count > 1 \
          else SuggestDecision([])
      return SuggestDecision([default] + list(rest.suggestions),
                             metadata=rest.metadata)
    return original_suggest(self, request)

  policy_cls.suggest = suggest
  return policy_cls
