"""Scalarization functions for multi-objective reduction.

Capability parity with vizier/_src/algorithms/designers/scalarization.py
(Linear/Chebyshev/HyperVolume/LinearAugmented scalarizers :40-114),
NumPy-based (the GPU path scalarizes inside the acquisition kernel).
"""

from __future__ import annotations

import abc
from typing import Optional

import numpy as np


class Scalarization(abc.ABC):
  """Maps (..., M) objective vectors to (...) scalars (maximize)."""

  def __init__(self, weights: np.ndarray):
    self.weights = np.asarray(weights, dtype=np.float64)
    if (self.weights <= 0).any():
      raise ValueError('Scalarization weights must be positive.')

  @abc.abstractmethod
  def __call__(self, objectives: np.ndarray) -> np.ndarray:
    ...


class LinearScalarization(Scalarization):

  def __call__(self, objectives: np.ndarray) -> np.ndarray:
    return np.asarray(objectives) @ self.weights


class ChebyshevScalarization(Scalarization):

  def __call__(self, objectives: np.ndarray) -> np.ndarray:
    return np.min(np.asarray(objectives) * self.weights, axis=-1)


class HyperVolumeScalarization(Scalarization):
  """min_m (y_m - ref_m)/w_m (arXiv:2006.04655 scalarization)."""

  def __init__(self, weights: np.ndarray,
               reference_point: Optional[np.ndarray] = None):
    super().__init__(weights)
    self.reference_point = (np.asarray(reference_point, dtype=np.float64)
                            if reference_point is not None else None)

  def __call__(self, objectives: np.ndarray) -> np.ndarray:
    ys = np.asarray(objectives, dtype=np.float64)
    if self.reference_point is not None:
      ys = ys - self.reference_point
    return np.min(ys / self.weights, axis=-1)


class LinearAugmentedScalarization(Scalarization):
  """Primary scalarizer + small linear sum augmentation."""

  def __init__(self, weights: np.ndarray,
               scalarization_factory=ChebyshevScalarization,
               augment_weight: float = 1.0):
    super().__init__(weights)
    self._primary = scalarization_factory(weights)
    self._augment_weight = augment_weight

  def __call__(self, objectives: np.ndarray) -> np.ndarray:
    return self._primary(objectives) + self._augment_weight * (
        np.asarray(objectives) @ self.weights)
