"""Gradient-free optimizer abstractions + conditional-space branching.

Capability parity with vizier/_src/algorithms/optimizers/base.py
(GradientFreeOptimizer, BranchSelector, BranchThenOptimizer).
"""

from __future__ import annotations

import abc
import dataclasses
from typing import Callable, Dict, List, Optional, Sequence

import numpy as np

from vizier_amd import pyvizier as vz

# Maps a batch of suggestions to their acquisition scores (maximize).
BatchTrialScoreFunction = Callable[[Sequence[vz.TrialSuggestion]],
                                   np.ndarray]


class GradientFreeOptimizer(abc.ABC):
  """Optimizes a score function over a search space."""

  @abc.abstractmethod
  def optimize(self, score_fn: BatchTrialScoreFunction,
               problem: vz.ProblemStatement, *, count: int = 1,
               seed_candidates: Sequence[vz.TrialSuggestion] = ()
               ) -> List[vz.TrialSuggestion]:
    ...


@dataclasses.dataclass(frozen=True)
class BranchSelection:
  """A flat subspace plus the number of suggestions to generate in it.

  N suggestions on a conditional space are decomposed into
  N_1 + ... + N_k suggestions over flat subspaces (base.py:50).
  """

  problem: vz.ProblemStatement
  num_suggestions: int


class BranchSelector(abc.ABC):
  """Chooses conditional-space branches to optimize within."""

  @abc.abstractmethod
  def select_branches(self, problem: vz.ProblemStatement, count: int
                      ) -> List[Dict[str, vz.ParameterValueTypes]]:
    """Returns parent-value assignments defining flat subproblems."""


class TopBranchSelector(BranchSelector):
  """Enumerates parent assignments in feasible-value order."""

  def select_branches(self, problem, count):
    parents = [pc for pc in problem.search_space.parameters
               if pc.child_parameter_configs]
    if not parents:
      return [{}]
    out = []
    parent = parents[0]
    for v in parent.feasible_values[:count]:
      out.append({parent.name: v})
    return out


class BranchThenOptimizer(GradientFreeOptimizer):
  """Optimizes each conditional branch with an inner optimizer."""

  def __init__(self, inner_factory: Callable[[], GradientFreeOptimizer],
               branch_selector: Optional[BranchSelector] = None,
               max_num_branches: int = 5):
    self._inner_factory = inner_factory
    self._selector = branch_selector or TopBranchSelector()
    self._max_num_branches = max_num_branches

  def optimize(self, score_fn, problem, *, count=1, seed_candidates=()):
    branches = self._selector.select_branches(problem,
                                              self._max_num_branches)
    candidates: List[vz.TrialSuggestion] = []
    scores: List[float] = []
    for branch in branches:
      sub_problem = _flatten_branch(problem, branch)
      inner = self._inner_factory()

      def branch_score(suggestions, _branch=branch):
        merged = []
        for s in suggestions:
          params = dict(s.parameters.as_dict())
          params.update(_branch)
          merged.append(vz.TrialSuggestion(params))
        return score_fn(merged)

      best = inner.optimize(branch_score, sub_problem, count=count)
      for s in best:
        params = dict(s.parameters.as_dict())
        params.update(branch)
        merged = vz.TrialSuggestion(params)
        candidates.append(merged)
        scores.append(float(score_fn([merged])[0]))
    order = np.argsort(-np.asarray(scores))
    return [candidates[i] for i in order[:count]]


def _flatten_branch(problem: vz.ProblemStatement,
                    branch: Dict[str, vz.ParameterValueTypes]
                    ) -> vz.ProblemStatement:
  """Search space of the children active under `branch` + flat params."""
  space = vz.SearchSpace()
  for pc in problem.search_space.parameters:
    if pc.name in branch:
      for sub in [pc.subspaces_by_value.get(pc.cast_value(
          branch[pc.name]))]:
        if sub is not None:
          for child in sub.parameters:
            space.add(child)
    else:
      space.add(pc.clone_without_children())
  return vz.ProblemStatement(search_space=space,
                             metric_information=list(
                                 problem.metric_information))
