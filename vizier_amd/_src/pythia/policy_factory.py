"""PolicyFactory protocol (parity with vizier/_src/pythia/policy_factory.py)."""

from __future__ import annotations

import abc

from vizier_amd._src.pyvizier.base_study_config import ProblemStatement
from vizier_amd._src.pythia.policy import Policy
from vizier_amd._src.pythia.policy_supporter import PolicySupporter


class PolicyFactory(abc.ABC):
  """Creates a Policy for a given study + algorithm string."""

  @abc.abstractmethod
  def __call__(self, problem_statement: ProblemStatement, algorithm: str,
               policy_supporter: PolicySupporter, study_name: str) -> Policy:
    ...
