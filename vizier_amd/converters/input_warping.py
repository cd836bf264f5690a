"""Kumaraswamy input warping (parity with
vizier/pyvizier/converters/input_warping.py)."""

from __future__ import annotations

import numpy as np


class KumaraswamyInputWarper:
  """Warps [0,1] features with the Kumaraswamy CDF: 1-(1-x^a)^b."""

  def __init__(self, a: float = 1.0, b: float = 1.0):
    if a <= 0 or b <= 0:
      raise ValueError('Kumaraswamy parameters must be positive.')
    self.a = a
    self.b = b

  def warp(self, features: np.ndarray) -> np.ndarray:
    x = np.clip(np.asarray(features, dtype=np.float64), 0.0, 1.0)
    return 1.0 - (1.0 - x ** self.a) ** self.b

  def unwarp(self, features: np.ndarray) -> np.ndarray:
    y = np.clip(np.asarray(features, dtype=np.float64), 0.0, 1.0)
    return (1.0 - (1.0 - y) ** (1.0 / self.b)) ** (1.0 / self.a)
