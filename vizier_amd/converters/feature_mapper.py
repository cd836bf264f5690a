"""Continuous/categorical feature splitting for model inputs.

Capability parity with vizier/pyvizier/converters/feature_mapper.py
(ContinuousCategoricalFeatureMapper :26): splits a TrialToArrayConverter
feature matrix into a continuous block and integer categorical indices
(one-hot -> argmax index per categorical parameter), and maps back
(index -> one-hot). NumPy-native; the GPU Eagle codec
(vizier_amd/_src/algorithms/optimizers/vectorized.py) is the torch
equivalent used on the hot path.
"""

from __future__ import annotations

import dataclasses
from typing import List

import numpy as np

from vizier_amd.converters.core import TrialToArrayConverter


@dataclasses.dataclass
class ContinuousAndCategoricalArray:
  continuous: np.ndarray    # (..., C) float
  categorical: np.ndarray   # (..., P) int32 category indices


class ContinuousCategoricalFeatureMapper:
  """Splits dense features by type (feature_mapper.py:26-170).

  Notation: C = continuous params, P = categorical params,
  F = total one-hot dimensions, B = batch.
  """

  def __init__(self, converter: TrialToArrayConverter):
    self._converter = converter
    self._continuous_indices: List[int] = []
    self._categorical_slices: List[slice] = []
    for col in converter.output_specs:
      if col.is_onehot:
        self._categorical_slices.append(
            slice(col.start, col.start + col.width))
      else:
        self._continuous_indices.append(col.start)
    self.n_categorical_params = len(self._categorical_slices)
    self.categorical_dims = [s.stop - s.start
                             for s in self._categorical_slices]

  def map(self, features: np.ndarray) -> ContinuousAndCategoricalArray:
    """(..., n_features) -> continuous block + integer category indices."""
    features = np.asarray(features)
    batch_shape = features.shape[:-1]
    continuous = features[..., self._continuous_indices]
    if self.n_categorical_params:
      categorical = np.stack(
          [features[..., s].argmax(axis=-1)
           for s in self._categorical_slices], axis=-1).astype(np.int32)
    else:
      categorical = np.zeros(batch_shape + (0,), dtype=np.int32)
    return ContinuousAndCategoricalArray(continuous=continuous,
                                         categorical=categorical)

  def unmap(self, features: ContinuousAndCategoricalArray) -> np.ndarray:
    """Inverse of map: rebuild the dense one-hot feature matrix."""
    cont = np.asarray(features.continuous)
    cat = np.asarray(features.categorical)
    if cont.shape[:-1] != cat.shape[:-1]:
      raise ValueError("'continuous' and 'categorical' batch shapes "
                       f'differ: {cont.shape[:-1]} vs {cat.shape[:-1]}')
    out = np.zeros(cont.shape[:-1] + (self._converter.n_features,),
                   dtype=cont.dtype if cont.size else np.float64)
    for i, idx in enumerate(self._continuous_indices):
      out[..., idx] = cont[..., i]
    for p, s in enumerate(self._categorical_slices):
      width = s.stop - s.start
      out[..., s] = np.eye(width, dtype=out.dtype)[cat[..., p]]
    return out
