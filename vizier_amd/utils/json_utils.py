"""NumPy-aware JSON encoding (parity with vizier/utils/json_utils.py)."""

from __future__ import annotations

import json
from typing import Any

import numpy as np


class NumpyEncoder(json.JSONEncoder):
  """Encodes numpy arrays/scalars into tagged JSON objects."""

  def default(self, o: Any):
    if isinstance(o, np.ndarray):
      return {'__numpy__': True, 'dtype': str(o.dtype),
              'shape': list(o.shape), 'data': o.ravel().tolist()}
    if isinstance(o, (np.integer,)):
      return int(o)
    if isinstance(o, (np.floating,)):
      return float(o)
    if isinstance(o, (np.bool_,)):
      return bool(o)
    return super().default(o)


def numpy_hook(obj: dict) -> Any:
  """json.loads object_hook decoding NumpyEncoder output."""
  if obj.get('__numpy__'):
    return np.asarray(obj['data'], dtype=obj['dtype']).reshape(
        obj['shape'])
  return obj


def dumps(obj: Any, **kwargs) -> str:
  return json.dumps(obj, cls=NumpyEncoder, **kwargs)


def loads(s: str, **kwargs) -> Any:
  return json.loads(s, object_hook=numpy_hook, **kwargs)
