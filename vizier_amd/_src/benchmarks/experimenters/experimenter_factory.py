"""Experimenter factories (parity with
vizier/_src/benchmarks/experimenters/experimenter_factory.py:73,110)."""

from __future__ import annotations

import dataclasses
from typing import Dict, Optional, Sequence

import numpy as np

from vizier_amd._src.benchmarks.experimenters import wrappers
from vizier_amd._src.benchmarks.experimenters.experimenter import (
    Experimenter,
)
from vizier_amd._src.benchmarks.experimenters.numpy_experimenter import (
    NumpyExperimenter,
)
from vizier_amd._src.benchmarks.experimenters.synthetic import bbob


@dataclasses.dataclass
class BBOBExperimenterFactory:
  """Creates a NumpyExperimenter for a named BBOB function."""

  name: str
  dim: int

  def __call__(self) -> Experimenter:
    fn = getattr(bbob, self.name, None)
    if fn is None:
      raise ValueError(f'Unknown BBOB function: {self.name}')
    return NumpyExperimenter(fn, bbob.DefaultBBOBProblemStatement(self.dim))


@dataclasses.dataclass
class SingleObjectiveExperimenterFactory:
  """Composes shift / noise / discretization over a base factory."""

  base_factory: BBOBExperimenterFactory
  shift: Optional[np.ndarray] = None
  noise_std: Optional[float] = None
  discrete_dict: Optional[Dict[str, Sequence[float]]] = None
  seed: int = 0

  def __call__(self) -> Experimenter:
    exptr = self.base_factory()
    if self.shift is not None:
      exptr = wrappers.ShiftingExperimenter(exptr, np.asarray(self.shift))
    if self.discrete_dict:
      exptr = wrappers.DiscretizingExperimenter(exptr, self.discrete_dict)
    if self.noise_std is not None:
      exptr = wrappers.NoisyExperimenter(exptr, noise_std=self.noise_std,
                                         seed=self.seed)
    return exptr
