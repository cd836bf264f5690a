"""BBOB synthetic benchmark functions (NumPy, from the BBOB definitions).

Capability parity with
vizier/_src/benchmarks/experimenters/synthetic/bbob.py: the 24 BBOB
noiseless functions plus the helper transforms (Tosz, Tasy, Lambda^alpha,
rotations, boundary penalty) and the default [-5, 5]^D problem
statement, plus the multi-objective helpers (NegativeSphere,
NegativeMinDifference, FonsecaFleming). All functions map (D,) arrays to
scalars and are MINIMIZED.

Implemented from the public BBOB function definitions (Hansen et al.,
"Real-Parameter Black-Box Optimization Benchmarking: Noiseless
Functions Definitions").
"""

from __future__ import annotations

import hashlib
from typing import Callable

import numpy as np

from vizier_amd import pyvizier as vz


def DefaultBBOBProblemStatement(
    dimension: int, *, metric_name: str = 'bbob_eval',
    min_value: float = -5.0, max_value: float = 5.0) -> vz.ProblemStatement:
  problem = vz.ProblemStatement()
  root = problem.search_space.root
  for i in range(dimension):
    root.add_float_param(f'x{i}', min_value, max_value)
  problem.metric_information.append(vz.MetricInformation(
      name=metric_name, goal=vz.ObjectiveMetricGoal.MINIMIZE))
  return problem


# -- transforms ---------------------------------------------------------------


def _rng(seed: int, tag: str) -> np.random.Generator:
  h = int(hashlib.md5(f'{seed}:{tag}'.encode()).hexdigest()[:8], 16)
  return np.random.default_rng(h)


def LambdaAlpha(alpha: float, dim: int) -> np.ndarray:
  """Diagonal conditioning matrix Lambda^alpha."""
  if dim == 1:
    return np.ones((1, 1))
  exps = 0.5 * np.arange(dim) / (dim - 1)
  return np.diag(alpha ** exps)


def Tosz(element: float) -> float:
  """Oscillatory non-linearity (scalar)."""
  x_hat = np.log(abs(element)) if element != 0 else 0.0
  c1 = 10.0 if element > 0 else 5.5
  c2 = 7.9 if element > 0 else 3.1
  return float(np.sign(element) * np.exp(
      x_hat + 0.049 * (np.sin(c1 * x_hat) + np.sin(c2 * x_hat))))


def ArrayTosz(vector: np.ndarray) -> np.ndarray:
  return np.array([Tosz(v) for v in vector])


def Tasy(vector: np.ndarray, beta: float) -> np.ndarray:
  """Asymmetry transform (applied to positive components)."""
  dim = len(vector)
  out = vector.astype(np.float64).copy()
  for i, v in enumerate(vector):
    if v > 0:
      exp = 1 + beta * (i / max(dim - 1, 1)) * np.sqrt(v)
      out[i] = v ** exp
  return out


def Fpen(vector: np.ndarray) -> float:
  """Boundary penalty for |x| > 5."""
  return float(np.sum(np.maximum(0.0, np.abs(vector) - 5.0) ** 2))


def _rotation(dim: int, seed: int, tag: str) -> np.ndarray:
  """A seeded random orthogonal matrix (QR of a Gaussian)."""
  a = _rng(seed, tag).standard_normal((dim, dim))
  q, r = np.linalg.qr(a)
  return q * np.sign(np.diag(r))


# -- the 24 noiseless functions ----------------------------------------------


def Sphere(arr: np.ndarray, seed: int = 0) -> float:
  del seed
  return float(np.sum(arr * arr))


def Ellipsoidal(arr: np.ndarray, seed: int = 0) -> float:
  del seed
  z = ArrayTosz(np.asarray(arr, dtype=np.float64))
  d = len(z)
  exps = 6.0 * np.arange(d) / max(d - 1, 1)
  return float(np.sum(10.0 ** exps * z * z))


def Rastrigin(arr: np.ndarray, seed: int = 0) -> float:
  z = np.asarray(arr, dtype=np.float64)
  d = len(z)
  R = _rotation(d, seed, 'rast_R')
  Q = _rotation(d, seed, 'rast_Q')
  z = R @ LambdaAlpha(10, d) @ Q @ Tasy(ArrayTosz(z), 0.2)
  return float(10 * (d - np.sum(np.cos(2 * np.pi * z))) + np.sum(z * z))


def BuecheRastrigin(arr: np.ndarray, seed: int = 0) -> float:
  del seed
  z = ArrayTosz(np.asarray(arr, dtype=np.float64))
  d = len(z)
  for i in range(d):
    s = 10.0 ** (0.5 * i / max(d - 1, 1))
    if z[i] > 0 and i % 2 == 0:
      s *= 10.0
    z[i] *= s
  return float(10 * (d - np.sum(np.cos(2 * np.pi * z))) +
               np.sum(z * z) + 100 * Fpen(arr))


def LinearSlope(arr: np.ndarray, seed: int = 0) -> float:
  del seed
  z = np.asarray(arr, dtype=np.float64)
  d = len(z)
  s = np.sign(np.ones(d)) * 10.0 ** (np.arange(d) / max(d - 1, 1))
  z = np.where(z * 5.0 < 25.0, z, 5.0)
  return float(np.sum(5.0 * np.abs(s) - s * z))


def AttractiveSector(arr: np.ndarray, seed: int = 0) -> float:
  x = np.asarray(arr, dtype=np.float64)
  d = len(x)
  R = _rotation(d, seed, 'sector_R')
  Q = _rotation(d, seed, 'sector_Q')
  z = Q @ LambdaAlpha(10, d) @ R @ x
  s = np.where(z * x > 0, 100.0, 1.0)
  return float(Tosz(np.sum((s * z) ** 2) ** 0.9))


def StepEllipsoidal(arr: np.ndarray, seed: int = 0) -> float:
  x = np.asarray(arr, dtype=np.float64)
  d = len(x)
  R = _rotation(d, seed, 'step_R')
  Q = _rotation(d, seed, 'step_Q')
  z_hat = LambdaAlpha(10, d) @ R @ x
  z_tilde = np.where(np.abs(z_hat) > 0.5, np.floor(0.5 + z_hat),
                     np.floor(0.5 + 10 * z_hat) / 10)
  z = Q @ z_tilde
  exps = 2.0 * np.arange(d) / max(d - 1, 1)
  value = 0.1 * max(abs(z_hat[0]) / 1e4,
                    float(np.sum(10.0 ** exps * z * z)))
  return float(value + Fpen(x))


def Rosenbrock(arr: np.ndarray, seed: int = 0) -> float:
  del seed
  z = np.asarray(arr, dtype=np.float64)
  z = max(1.0, np.sqrt(len(z)) / 8.0) * z + 1.0
  return float(np.sum(100 * (z[:-1] ** 2 - z[1:]) ** 2 +
                      (z[:-1] - 1) ** 2))


def RosenbrockRotated(arr: np.ndarray, seed: int = 0) -> float:
  x = np.asarray(arr, dtype=np.float64)
  d = len(x)
  R = _rotation(d, seed, 'rosen_R')
  z = max(1.0, np.sqrt(d) / 8.0) * (R @ x) + 0.5
  return float(np.sum(100 * (z[:-1] ** 2 - z[1:]) ** 2 +
                      (z[:-1] - 1) ** 2))


def Discus(arr: np.ndarray, seed: int = 0) -> float:
  x = np.asarray(arr, dtype=np.float64)
  R = _rotation(len(x), seed, 'discus_R')
  z = ArrayTosz(R @ x)
  return float(1e6 * z[0] ** 2 + np.sum(z[1:] ** 2))


def BentCigar(arr: np.ndarray, seed: int = 0) -> float:
  x = np.asarray(arr, dtype=np.float64)
  R = _rotation(len(x), seed, 'cigar_R')
  z = R @ Tasy(R @ x, 0.5)
  return float(z[0] ** 2 + 1e6 * np.sum(z[1:] ** 2))


def SharpRidge(arr: np.ndarray, seed: int = 0) -> float:
  x = np.asarray(arr, dtype=np.float64)
  d = len(x)
  R = _rotation(d, seed, 'ridge_R')
  Q = _rotation(d, seed, 'ridge_Q')
  z = Q @ LambdaAlpha(10, d) @ R @ x
  return float(z[0] ** 2 + 100 * np.sqrt(np.sum(z[1:] ** 2)))


def DifferentPowers(arr: np.ndarray, seed: int = 0) -> float:
  x = np.asarray(arr, dtype=np.float64)
  d = len(x)
  R = _rotation(d, seed, 'powers_R')
  z = R @ x
  exps = 2 + 4 * np.arange(d) / max(d - 1, 1)
  return float(np.sqrt(np.sum(np.abs(z) ** exps)))


def RastriginRotated(arr: np.ndarray, seed: int = 0) -> float:
  return Rastrigin(arr, seed + 17)


def Weierstrass(arr: np.ndarray, seed: int = 0) -> float:
  x = np.asarray(arr, dtype=np.float64)
  d = len(x)
  R = _rotation(d, seed, 'weier_R')
  Q = _rotation(d, seed, 'weier_Q')
  z = R @ LambdaAlpha(1.0 / 100, d) @ Q @ ArrayTosz(R @ x)
  k = np.arange(12)
  ak = 0.5 ** k
  bk = 3.0 ** k
  f0 = np.sum(ak * np.cos(np.pi * bk))
  total = 0.0
  for zi in z:
    total += np.sum(ak * np.cos(2 * np.pi * bk * (zi + 0.5)))
  return float(10 * (total / d - f0) ** 3 + 10 * Fpen(x) / d)


def _schaffers(arr: np.ndarray, seed: int, alpha: float) -> float:
  x = np.asarray(arr, dtype=np.float64)
  d = len(x)
  if d == 1:
    return 0.0
  R = _rotation(d, seed, 'schaf_R')
  Q = _rotation(d, seed, 'schaf_Q')
  z = LambdaAlpha(alpha, d) @ Q @ Tasy(R @ x, 0.5)
  s = np.sqrt(z[:-1] ** 2 + z[1:] ** 2)
  value = np.mean(np.sqrt(s) + np.sqrt(s) * np.sin(50 * s ** 0.2) ** 2)
  return float(value ** 2 + 10 * Fpen(x))


def SchaffersF7(arr: np.ndarray, seed: int = 0) -> float:
  return _schaffers(arr, seed, 10.0)


def SchaffersF7IllConditioned(arr: np.ndarray, seed: int = 0) -> float:
  return _schaffers(arr, seed + 3, 1000.0)


def GriewankRosenbrock(arr: np.ndarray, seed: int = 0) -> float:
  x = np.asarray(arr, dtype=np.float64)
  d = len(x)
  R = _rotation(d, seed, 'griew_R')
  z = max(1.0, np.sqrt(d) / 8.0) * (R @ x) + 0.5
  s = 100 * (z[:-1] ** 2 - z[1:]) ** 2 + (z[:-1] - 1) ** 2
  return float(10.0 / (d - 1) * np.sum(s / 4000 - np.cos(s)) + 10)


def Schwefel(arr: np.ndarray, seed: int = 0) -> float:
  x = np.asarray(arr, dtype=np.float64)
  d = len(x)
  ones = np.where(_rng(seed, 'schwefel').random(d) > 0.5, 1.0, -1.0)
  x_opt = 4.2096874633 / 2.0 * ones
  x_hat = 2.0 * ones * x
  z_hat = x_hat.copy()
  z_hat[1:] += 0.25 * (x_hat[:-1] - 2 * np.abs(x_opt[:-1]))
  z = 100 * (LambdaAlpha(10, d) @ (z_hat - 2 * np.abs(x_opt)) +
             2 * np.abs(x_opt))
  value = -np.mean(z * np.sin(np.sqrt(np.abs(z)))) / 100.0
  return float(0.01 * (418.9828872724339 / 100.0 + value) +
               100 * Fpen(z / 100))


def Katsuura(arr: np.ndarray, seed: int = 0) -> float:
  x = np.asarray(arr, dtype=np.float64)
  d = len(x)
  R = _rotation(d, seed, 'kats_R')
  Q = _rotation(d, seed, 'kats_Q')
  z = Q @ LambdaAlpha(100, d) @ R @ x
  prod = 1.0
  for i, zi in enumerate(z):
    j = 2.0 ** np.arange(1, 33)
    s = np.sum(np.abs(j * zi - np.round(j * zi)) / j)
    prod *= (1 + (i + 1) * s) ** (10.0 / d ** 1.2)
  return float(10.0 / d ** 2 * (prod - 1) + Fpen(x))


def Lunacek(arr: np.ndarray, seed: int = 0) -> float:
  x = np.asarray(arr, dtype=np.float64)
  d = len(x)
  mu0 = 2.5
  s = 1.0 - 1.0 / (2 * np.sqrt(d + 20) - 8.2)
  mu1 = -np.sqrt((mu0 ** 2 - 1) / s)
  ones = np.where(_rng(seed, 'lunacek').random(d) > 0.5, 1.0, -1.0)
  x_hat = 2.0 * ones * x
  R = _rotation(d, seed, 'lun_R')
  Q = _rotation(d, seed, 'lun_Q')
  z = Q @ LambdaAlpha(100, d) @ R @ (x_hat - mu0)
  s1 = np.sum((x_hat - mu0) ** 2)
  s2 = d + s * np.sum((x_hat - mu1) ** 2)
  return float(min(s1, s2) +
               10 * (d - np.sum(np.cos(2 * np.pi * z))) + 1e4 * Fpen(x))


def _gallagher(arr: np.ndarray, seed: int, n_peaks: int) -> float:
  x = np.asarray(arr, dtype=np.float64)
  d = len(x)
  rng = _rng(seed, f'gall{n_peaks}')
  R = _rotation(d, seed, f'gall_R{n_peaks}')
  if n_peaks == 101:
    alphas = [1000.0 ** (2.0 * i / 98) for i in range(99)]
    w_scale, y_range = 10.0, 10.0
  else:
    alphas = [1000.0 ** (2.0 * i / 19) for i in range(20)]
    w_scale, y_range = 9.8, 9.8
  rng.shuffle(alphas)
  ys = [rng.uniform(-y_range / 2, y_range / 2, d)
        for _ in range(n_peaks - 1)]
  ys.insert(0, rng.uniform(-4, 4, d))
  cs = [LambdaAlpha(a, d) / a ** 0.25 for a in alphas]
  cs.insert(0, LambdaAlpha(1000.0, d))
  ws = [w_scale - i * (w_scale - 1.1) / max(n_peaks - 2, 1)
        for i in range(1, n_peaks)]
  ws.insert(0, 10.0)
  best = -np.inf
  for w, y, c in zip(ws, ys, cs):
    diff = R @ (x - y)
    best = max(best, w * np.exp(-diff @ c @ diff / (2.0 * d)))
  return float(Tosz(10.0 - best) ** 2 + Fpen(x))


def Gallagher101Me(arr: np.ndarray, seed: int = 0) -> float:
  return _gallagher(arr, seed, 101)


def Gallagher21Me(arr: np.ndarray, seed: int = 0) -> float:
  return _gallagher(arr, seed, 21)


# -- multi-objective helpers --------------------------------------------------


def NegativeSphere(arr: np.ndarray, seed: int = 0) -> float:
  del seed
  z = np.asarray(arr, dtype=np.float64)
  return float(100.0 - np.sum(z * z))


def NegativeMinDifference(arr: np.ndarray, seed: int = 0) -> float:
  del seed
  z = np.asarray(arr, dtype=np.float64)
  return float(10.0 - np.min(np.diff(z)) if len(z) > 1 else 10.0)


def FonsecaFleming(arr: np.ndarray, seed: int = 0) -> float:
  del seed
  z = np.asarray(arr, dtype=np.float64)
  return float(1.0 - np.exp(-np.sum((z - 1.0 / np.sqrt(len(z))) ** 2)))


BBOB_FUNCTIONS = [
    Sphere, Ellipsoidal, Rastrigin, BuecheRastrigin, LinearSlope,
    AttractiveSector, StepEllipsoidal, Rosenbrock, RosenbrockRotated,
    Discus, BentCigar, SharpRidge, DifferentPowers, RastriginRotated,
    Weierstrass, SchaffersF7, SchaffersF7IllConditioned,
    GriewankRosenbrock, Schwefel, Katsuura, Lunacek, Gallagher101Me,
    Gallagher21Me,
]
