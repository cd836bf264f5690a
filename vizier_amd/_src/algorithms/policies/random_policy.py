"""RandomPolicy (parity with
vizier/_src/algorithms/policies/random_policy.py:29)."""

from __future__ import annotations

from typing import Optional

from vizier_amd._src.pythia.policy import (
    EarlyStopDecision,
    EarlyStopDecisions,
    EarlyStopRequest,
    Policy,
    SuggestDecision,
    SuggestRequest,
)
from vizier_amd._src.pythia.policy_supporter import PolicySupporter
from vizier_amd._src.algorithms.designers.random import RandomDesigner


class RandomPolicy(Policy):
  """Suggests uniformly random trials; random early stopping."""

  def __init__(self, policy_supporter: PolicySupporter,
               seed: Optional[int] = None):
    self._supporter = policy_supporter
    self._seed = seed

  def suggest(self, request: SuggestRequest) -> SuggestDecision:
    designer = RandomDesigner(request.study_config.search_space,
                              seed=self._seed)
    return SuggestDecision(designer.suggest(request.count))

  def early_stop(self, request: EarlyStopRequest) -> EarlyStopDecisions:
    decisions = [EarlyStopDecision(id=tid, reason='Random policy never stops',
                                   should_stop=False)
                 for tid in request.trial_ids]
    return EarlyStopDecisions(decisions=decisions)
