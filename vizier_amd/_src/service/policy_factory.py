"""Algorithm-string -> Policy registry.

Capability parity with vizier/_src/service/policy_factory.py:28-115.
Imports are lazy so that e.g. the GP stack (torch + HIP kernels) is only
loaded when a GP algorithm is requested.
"""

from __future__ import annotations

from vizier_amd._src.pyvizier.base_study_config import ProblemStatement
from vizier_amd._src.pythia.policy import Policy
from vizier_amd._src.pythia.policy_factory import PolicyFactory
from vizier_amd._src.pythia.policy_supporter import PolicySupporter


class DefaultPolicyFactory(PolicyFactory):
  """Maps the built-in algorithm strings to policies."""

  def __call__(self, problem_statement: ProblemStatement, algorithm: str,
               policy_supporter: PolicySupporter, study_name: str) -> Policy:
    del study_name
    algorithm = algorithm or 'ALGORITHM_UNSPECIFIED'
    if algorithm in ('ALGORITHM_UNSPECIFIED', 'DEFAULT', 'GP_UCB_PE'):
      from vizier_amd._src.algorithms.designers import gp_ucb_pe
      from vizier_amd._src.algorithms.policies import designer_policy as dp
      return dp.DesignerPolicy(policy_supporter,
                               gp_ucb_pe.VizierGPUCBPEBandit.from_problem)
    if algorithm == 'GAUSSIAN_PROCESS_BANDIT':
      from vizier_amd._src.algorithms.designers import gp_bandit
      from vizier_amd._src.algorithms.policies import designer_policy as dp
      return dp.DesignerPolicy(policy_supporter,
                               gp_bandit.VizierGPBandit.from_problem)
    if algorithm == 'RANDOM_SEARCH':
      from vizier_amd._src.algorithms.policies import random_policy
      return random_policy.RandomPolicy(policy_supporter)
    if algorithm == 'QUASI_RANDOM_SEARCH':
      from vizier_amd._src.algorithms.designers import quasi_random
      from vizier_amd._src.algorithms.policies import designer_policy as dp
      return dp.PartiallySerializableDesignerPolicy(
          problem_statement, policy_supporter,
          quasi_random.QuasiRandomDesigner.from_problem, ns_root='quasirandom')
    if algorithm == 'GRID_SEARCH':
      from vizier_amd._src.algorithms.designers import grid
      from vizier_amd._src.algorithms.policies import designer_policy as dp
      return dp.PartiallySerializableDesignerPolicy(
          problem_statement, policy_supporter,
          grid.GridSearchDesigner.from_problem, ns_root='grid')
    if algorithm == 'SHUFFLED_GRID_SEARCH':
      from vizier_amd._src.algorithms.designers import grid
      from vizier_amd._src.algorithms.policies import designer_policy as dp

      def factory(problem, _seed=875):
        return grid.GridSearchDesigner.from_problem(problem,
                                                    shuffle_seed=_seed)
      return dp.PartiallySerializableDesignerPolicy(
          problem_statement, policy_supporter, factory, ns_root='shuffled')
    if algorithm == 'NSGA2':
      from vizier_amd._src.algorithms.evolution import nsga2
      from vizier_amd._src.algorithms.policies import designer_policy as dp
      return dp.DesignerPolicy(policy_supporter, nsga2.NSGA2Designer)
    if algorithm == 'EAGLE_STRATEGY':
      from vizier_amd._src.algorithms.designers.eagle_strategy import (
          eagle_strategy,
      )
      from vizier_amd._src.algorithms.policies import designer_policy as dp
      return dp.DesignerPolicy(policy_supporter,
                               eagle_strategy.EagleStrategyDesigner)
    if algorithm == 'CMA_ES':
      from vizier_amd._src.algorithms.designers import cmaes
      from vizier_amd._src.algorithms.policies import designer_policy as dp
      return dp.DesignerPolicy(policy_supporter, cmaes.CMAESDesigner)
    if algorithm == 'BOCS':
      from vizier_amd._src.algorithms.designers import bocs
      from vizier_amd._src.algorithms.policies import designer_policy as dp
      return dp.DesignerPolicy(policy_supporter, bocs.BOCSDesigner)
    if algorithm == 'HARMONICA':
      from vizier_amd._src.algorithms.designers import harmonica
      from vizier_amd._src.algorithms.policies import designer_policy as dp
      return dp.DesignerPolicy(policy_supporter, harmonica.HarmonicaDesigner)
    raise ValueError(f'Unknown algorithm: {algorithm}')
