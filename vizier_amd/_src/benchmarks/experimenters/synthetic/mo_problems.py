"""Native DTLZ / ZDT / WFG multi-objective benchmark suites.

Capability parity with
vizier/_src/benchmarks/experimenters/synthetic/multiobjective_optproblems.py
(DTLZExperimenterFactory :119, WFGExperimenterFactory :59,
ZDTExperimenterFactory :174). The reference delegates the function
bodies to the `optproblems` PyPI package; that package is not available
offline, so the canonical formulas (Deb et al. 2005 for DTLZ; Zitzler,
Deb & Thiele 2000 for ZDT; Huband et al. 2006 for WFG) are implemented
here directly in NumPy.

All three factories expose [0, 1]^dim problem statements like the
reference does. ZDT4 and WFG internally rescale to their canonical
domains (x_i in [-5, 5] for ZDT4 distance parameters, z_i in [0, 2i]
for WFG) so the normalized search space still covers the full problem.
"""

from __future__ import annotations

import dataclasses
import json
import math
from typing import Callable, List, Sequence

import numpy as np

from vizier_amd import pyvizier as vz
from vizier_amd._src.benchmarks.experimenters.extra import (
    MultiObjectiveNumpyExperimenter,
    SerializableExperimenterFactory,
)

EXPERIMENTER_FACTORY_KEY = 'experimenter_factory'


def _mo_problem_statement(dimension: int, num_objectives: int,
                          ) -> vz.ProblemStatement:
  """[0,1]^dim, num_objectives MINIMIZE metrics f0..f{M-1}."""
  problem = vz.ProblemStatement()
  for n in range(num_objectives):
    problem.metric_information.append(vz.MetricInformation(
        name=f'f{n}', goal=vz.ObjectiveMetricGoal.MINIMIZE))
  for d in range(dimension):
    problem.search_space.root.add_float_param(f'x{d}', 0.0, 1.0)
  return problem


# ---------------------------------------------------------------------------
# DTLZ (Deb, Thiele, Laumanns, Zitzler 2005). x split into M-1 position
# parameters and n-M+1 distance parameters x_M.
# ---------------------------------------------------------------------------


def _dtlz_g1(xm: np.ndarray) -> float:
  return 100.0 * (xm.size + float(np.sum(
      (xm - 0.5) ** 2 - np.cos(20.0 * np.pi * (xm - 0.5)))))


def _dtlz_g2(xm: np.ndarray) -> float:
  return float(np.sum((xm - 0.5) ** 2))


def _dtlz_linear_shape(pos: np.ndarray, g: float) -> np.ndarray:
  """DTLZ1 shape: f_m = 0.5 (1+g) prod(pos_i) * (1 - pos_cut)."""
  m_objs = pos.size + 1
  f = np.empty(m_objs)
  for m in range(m_objs):
    val = 0.5 * (1.0 + g)
    val *= float(np.prod(pos[:m_objs - 1 - m]))
    if m > 0:
      val *= 1.0 - pos[m_objs - 1 - m]
    f[m] = val
  return f


def _dtlz_spherical_shape(theta: np.ndarray, g: float) -> np.ndarray:
  """DTLZ2 shape over angles theta in [0, pi/2]."""
  m_objs = theta.size + 1
  f = np.empty(m_objs)
  for m in range(m_objs):
    val = 1.0 + g
    val *= float(np.prod(np.cos(theta[:m_objs - 1 - m])))
    if m > 0:
      val *= math.sin(theta[m_objs - 1 - m])
    f[m] = val
  return f


def _dtlz(name: str, num_objectives: int,
          ) -> Callable[[np.ndarray], Sequence[float]]:
  """Returns x -> objective tuple for DTLZ1..DTLZ7."""
  m_objs = num_objectives

  def impl(x: np.ndarray) -> Sequence[float]:
    x = np.asarray(x, dtype=np.float64).reshape(-1)
    pos, xm = x[:m_objs - 1], x[m_objs - 1:]
    if name == 'DTLZ1':
      return _dtlz_linear_shape(pos, _dtlz_g1(xm))
    if name == 'DTLZ2':
      return _dtlz_spherical_shape(pos * (np.pi / 2), _dtlz_g2(xm))
    if name == 'DTLZ3':
      return _dtlz_spherical_shape(pos * (np.pi / 2), _dtlz_g1(xm))
    if name == 'DTLZ4':
      return _dtlz_spherical_shape((pos ** 100) * (np.pi / 2), _dtlz_g2(xm))
    if name in ('DTLZ5', 'DTLZ6'):
      g = (float(np.sum(xm ** 0.1)) if name == 'DTLZ6' else _dtlz_g2(xm))
      theta = np.empty_like(pos)
      if pos.size:
        theta[0] = pos[0] * (np.pi / 2)
        theta[1:] = (np.pi / (4.0 * (1.0 + g))) * (1.0 + 2.0 * g * pos[1:])
      return _dtlz_spherical_shape(theta, g)
    if name == 'DTLZ7':
      f = np.empty(m_objs)
      f[:m_objs - 1] = pos
      g = 1.0 + 9.0 * float(np.mean(xm)) if xm.size else 1.0
      h = m_objs - float(np.sum(
          f[:m_objs - 1] / (1.0 + g)
          * (1.0 + np.sin(3.0 * np.pi * f[:m_objs - 1]))))
      f[m_objs - 1] = (1.0 + g) * h
      return f
    raise ValueError(f'{name} is not a valid DTLZ problem')

  return impl


# ---------------------------------------------------------------------------
# ZDT (Zitzler, Deb, Thiele 2000). Bi-objective, first var is position.
# ---------------------------------------------------------------------------


def _zdt(name: str) -> Callable[[np.ndarray], Sequence[float]]:
  def impl(x: np.ndarray) -> Sequence[float]:
    x = np.asarray(x, dtype=np.float64).reshape(-1)
    n = x.size
    rest = x[1:]
    if name == 'ZDT1':
      f1 = x[0]
      g = 1.0 + 9.0 * float(np.sum(rest)) / max(n - 1, 1)
      return (f1, g * (1.0 - math.sqrt(f1 / g)))
    if name == 'ZDT2':
      f1 = x[0]
      g = 1.0 + 9.0 * float(np.sum(rest)) / max(n - 1, 1)
      return (f1, g * (1.0 - (f1 / g) ** 2))
    if name == 'ZDT3':
      f1 = x[0]
      g = 1.0 + 9.0 * float(np.sum(rest)) / max(n - 1, 1)
      return (f1, g * (1.0 - math.sqrt(f1 / g)
                       - (f1 / g) * math.sin(10.0 * np.pi * f1)))
    if name == 'ZDT4':
      f1 = x[0]
      scaled = rest * 10.0 - 5.0  # canonical x_i in [-5, 5]
      g = 1.0 + 10.0 * (n - 1) + float(np.sum(
          scaled ** 2 - 10.0 * np.cos(4.0 * np.pi * scaled)))
      return (f1, g * (1.0 - math.sqrt(f1 / g)))
    if name == 'ZDT6':
      f1 = 1.0 - math.exp(-4.0 * x[0]) * math.sin(6.0 * np.pi * x[0]) ** 6
      g = 1.0 + 9.0 * (float(np.sum(rest)) / max(n - 1, 1)) ** 0.25
      return (f1, g * (1.0 - (f1 / g) ** 2))
    raise ValueError(f'{name} is not a valid ZDT problem '
                     '(ZDT5 is bitstring-only and not supported, '
                     'matching the reference)')

  return impl


# ---------------------------------------------------------------------------
# WFG toolkit (Huband, Hingston, Barone, While 2006). Composition of
# bias/shift/reduction transformations followed by a shape function.
# All helpers operate on y in [0, 1].
# ---------------------------------------------------------------------------


def _b_poly(y: np.ndarray, alpha: float) -> np.ndarray:
  return y ** alpha


def _b_flat(y: np.ndarray, a: float, b: float, c: float) -> np.ndarray:
  out = (a + np.minimum(0.0, np.floor(y - b)) * (a * (b - y) / b)
         - np.minimum(0.0, np.floor(c - y)) * ((1.0 - a) * (y - c)
                                               / (1.0 - c)))
  return np.clip(out, 0.0, 1.0)


def _b_param(y: np.ndarray, u: np.ndarray, a: float, b: float, c: float
             ) -> np.ndarray:
  v = a - (1.0 - 2.0 * u) * np.abs(np.floor(0.5 - u) + a)
  return y ** (b + (c - b) * v)


def _s_linear(y: np.ndarray, a: float) -> np.ndarray:
  return np.abs(y - a) / np.abs(np.floor(a - y) + a)


def _s_decept(y: np.ndarray, a: float, b: float, c: float) -> np.ndarray:
  t1 = np.floor(y - a + b) * (1.0 - c + (a - b) / b) / (a - b)
  t2 = np.floor(a + b - y) * (1.0 - c + (1.0 - a - b) / b) / (1.0 - a - b)
  return 1.0 + (np.abs(y - a) - b) * (t1 + t2 + 1.0 / b)


def _s_multi(y: np.ndarray, a: float, b: float, c: float) -> np.ndarray:
  t1 = np.abs(y - c) / (2.0 * (np.floor(c - y) + c))
  t2 = (4.0 * a + 2.0) * np.pi * (0.5 - t1)
  return (1.0 + np.cos(t2) + 4.0 * b * t1 ** 2) / (b + 2.0)


def _r_sum(y: np.ndarray, w: np.ndarray) -> float:
  return float(np.sum(w * y) / np.sum(w))


def _r_nonsep(y: np.ndarray, a: int) -> float:
  n = y.size
  num = 0.0
  for j in range(n):
    num += y[j]
    for k in range(a - 1):
      num += abs(y[j] - y[(j + 1 + k) % n])
  denom = (n / a) * math.ceil(a / 2.0) * (1.0 + 2.0 * a
                                          - 2.0 * math.ceil(a / 2.0))
  return num / denom


def _wfg_shape_linear(x: np.ndarray, m: int, m_objs: int) -> float:
  # m is 1-based objective index.
  if m == 1:
    return float(np.prod(x[:m_objs - 1]))
  if m < m_objs:
    return float(np.prod(x[:m_objs - m])) * (1.0 - x[m_objs - m])
  return 1.0 - x[0]


def _wfg_shape_convex(x: np.ndarray, m: int, m_objs: int) -> float:
  c = 1.0 - np.cos(x * np.pi / 2.0)
  if m == 1:
    return float(np.prod(c[:m_objs - 1]))
  if m < m_objs:
    return float(np.prod(c[:m_objs - m])) * (
        1.0 - math.sin(x[m_objs - m] * np.pi / 2.0))
  return 1.0 - math.sin(x[0] * np.pi / 2.0)


def _wfg_shape_concave(x: np.ndarray, m: int, m_objs: int) -> float:
  s = np.sin(x * np.pi / 2.0)
  if m == 1:
    return float(np.prod(s[:m_objs - 1]))
  if m < m_objs:
    return float(np.prod(s[:m_objs - m])) * math.cos(
        x[m_objs - m] * np.pi / 2.0)
  return math.cos(x[0] * np.pi / 2.0)


def _wfg_shape_mixed(x: np.ndarray, alpha: float, a: float) -> float:
  aa = 2.0 * a * np.pi
  return (1.0 - x[0] - math.cos(aa * x[0] + np.pi / 2.0) / aa) ** alpha


def _wfg_shape_disc(x: np.ndarray, alpha: float, beta: float, a: float
                    ) -> float:
  return 1.0 - (x[0] ** alpha) * math.cos(a * (x[0] ** beta) * np.pi) ** 2


def _wfg_reduce_groups(y: np.ndarray, k: int, m_objs: int,
                       weighted: bool) -> np.ndarray:
  """Standard final reduction: k position params -> M-1 values via
  r_sum over contiguous groups, distance params -> one value."""
  t = np.empty(m_objs)
  group = k // (m_objs - 1)
  for m in range(m_objs - 1):
    lo, hi = m * group, (m + 1) * group
    w = (2.0 * np.arange(lo + 1, hi + 1) if weighted
         else np.ones(hi - lo))
    t[m] = _r_sum(y[lo:hi], w)
  w_dist = (2.0 * np.arange(k + 1, y.size + 1) if weighted
            else np.ones(y.size - k))
  t[m_objs - 1] = _r_sum(y[k:], w_dist)
  return t


def _wfg(name: str, m_objs: int, n: int, k: int,
         ) -> Callable[[np.ndarray], Sequence[float]]:
  """Returns x01 -> objectives for WFG1..WFG9; x01 in [0,1]^n."""
  l = n - k
  s_consts = 2.0 * np.arange(1, m_objs + 1)
  a_consts = np.ones(m_objs - 1)
  if name == 'WFG3':
    a_consts = np.zeros(m_objs - 1)
    a_consts[0] = 1.0

  def impl(x01: np.ndarray) -> Sequence[float]:
    # Canonical domain z_i in [0, 2i]; after normalization y = z/(2i)
    # the [0,1] inputs ARE y, so no explicit rescale is needed.
    y = np.clip(np.asarray(x01, dtype=np.float64).reshape(-1), 0.0, 1.0)

    if name == 'WFG1':
      y = y.copy()
      y[k:] = _s_linear(y[k:], 0.35)
      y[k:] = _b_flat(y[k:], 0.8, 0.75, 0.85)
      y = _b_poly(y, 0.02)
      t = _wfg_reduce_groups(y, k, m_objs, weighted=True)
    elif name in ('WFG2', 'WFG3'):
      y = y.copy()
      y[k:] = _s_linear(y[k:], 0.35)
      half = l // 2
      y2 = np.empty(k + half)
      y2[:k] = y[:k]
      for i in range(half):
        pair = y[k + 2 * i:k + 2 * i + 2]
        y2[k + i] = _r_nonsep(pair, 2)
      t = _wfg_reduce_groups(y2, k, m_objs, weighted=False)
    elif name == 'WFG4':
      y = _s_multi(y, 30.0, 10.0, 0.35)
      t = _wfg_reduce_groups(y, k, m_objs, weighted=False)
    elif name == 'WFG5':
      y = _s_decept(y, 0.35, 0.001, 0.05)
      t = _wfg_reduce_groups(y, k, m_objs, weighted=False)
    elif name == 'WFG6':
      y = y.copy()
      y[k:] = _s_linear(y[k:], 0.35)
      t = np.empty(m_objs)
      group = k // (m_objs - 1)
      for m in range(m_objs - 1):
        t[m] = _r_nonsep(y[m * group:(m + 1) * group], group)
      t[m_objs - 1] = _r_nonsep(y[k:], l)
    elif name == 'WFG7':
      y = y.copy()
      for i in range(k):
        u = _r_sum(y[i + 1:], np.ones(n - i - 1))
        y[i] = _b_param(y[i:i + 1], np.array([u]),
                        0.98 / 49.98, 0.02, 50.0)[0]
      y[k:] = _s_linear(y[k:], 0.35)
      t = _wfg_reduce_groups(y, k, m_objs, weighted=False)
    elif name == 'WFG8':
      y = y.copy()
      for i in range(k, n):
        u = _r_sum(y[:i], np.ones(i))
        y[i] = _b_param(y[i:i + 1], np.array([u]),
                        0.98 / 49.98, 0.02, 50.0)[0]
      y[k:] = _s_linear(y[k:], 0.35)
      t = _wfg_reduce_groups(y, k, m_objs, weighted=False)
    elif name == 'WFG9':
      y = y.copy()
      for i in range(n - 1):
        u = _r_sum(y[i + 1:], np.ones(n - i - 1))
        y[i] = _b_param(y[i:i + 1], np.array([u]),
                        0.98 / 49.98, 0.02, 50.0)[0]
      y[:k] = _s_decept(y[:k], 0.35, 0.001, 0.05)
      y[k:] = _s_multi(y[k:], 30.0, 95.0, 0.35)
      t = np.empty(m_objs)
      group = k // (m_objs - 1)
      for m in range(m_objs - 1):
        t[m] = _r_nonsep(y[m * group:(m + 1) * group], group)
      t[m_objs - 1] = _r_nonsep(y[k:], l)
    else:
      raise ValueError(f'{name} is not a valid WFG problem')

    x = np.empty(m_objs)
    x[:m_objs - 1] = (np.maximum(t[m_objs - 1], a_consts)
                      * (t[:m_objs - 1] - 0.5) + 0.5)
    x[m_objs - 1] = t[m_objs - 1]

    f = np.empty(m_objs)
    for m in range(1, m_objs + 1):
      if name == 'WFG1':
        h = (_wfg_shape_convex(x, m, m_objs) if m < m_objs
             else _wfg_shape_mixed(x, 1.0, 5.0))
      elif name == 'WFG2':
        h = (_wfg_shape_convex(x, m, m_objs) if m < m_objs
             else _wfg_shape_disc(x, 1.0, 1.0, 5.0))
      elif name == 'WFG3':
        h = _wfg_shape_linear(x, m, m_objs)
      else:
        h = _wfg_shape_concave(x, m, m_objs)
      f[m - 1] = x[m_objs - 1] + s_consts[m - 1] * h
    return f

  return impl


# ---------------------------------------------------------------------------
# Factories (dump/recover parity with the reference factories).
# ---------------------------------------------------------------------------


@dataclasses.dataclass
class DTLZExperimenterFactory(SerializableExperimenterFactory):
  """DTLZ1..DTLZ7 with `dim` variables and `num_objectives` metrics."""

  name: str = ''
  dim: int = 1
  num_objectives: int = 2

  def __post_init__(self):
    if self.dim < self.num_objectives:
      raise ValueError(
          f'DTLZ needs dim >= num_objectives, got {self.dim} < '
          f'{self.num_objectives}')

  def __call__(self) -> MultiObjectiveNumpyExperimenter:
    impl = _dtlz(self.name, self.num_objectives)
    problem = _mo_problem_statement(self.dim, self.num_objectives)
    return MultiObjectiveNumpyExperimenter(impl, problem)

  def dump(self) -> vz.Metadata:
    md = vz.Metadata()
    md[EXPERIMENTER_FACTORY_KEY] = json.dumps({
        'name': self.name, 'dim': self.dim,
        'num_objectives': self.num_objectives})
    return md

  @classmethod
  def recover(cls, metadata: vz.Metadata) -> 'DTLZExperimenterFactory':
    return cls(**json.loads(metadata[EXPERIMENTER_FACTORY_KEY]))


@dataclasses.dataclass
class ZDTExperimenterFactory(SerializableExperimenterFactory):
  """ZDT1/2/3/4/6 with `dim` variables (always bi-objective)."""

  name: str = ''
  dim: int = 1

  def __call__(self) -> MultiObjectiveNumpyExperimenter:
    impl = _zdt(self.name)
    problem = _mo_problem_statement(self.dim, 2)
    return MultiObjectiveNumpyExperimenter(impl, problem)

  def dump(self) -> vz.Metadata:
    md = vz.Metadata()
    md[EXPERIMENTER_FACTORY_KEY] = json.dumps({
        'name': self.name, 'dim': self.dim})
    return md

  @classmethod
  def recover(cls, metadata: vz.Metadata) -> 'ZDTExperimenterFactory':
    return cls(**json.loads(metadata[EXPERIMENTER_FACTORY_KEY]))


@dataclasses.dataclass
class WFGExperimenterFactory(SerializableExperimenterFactory):
  """WFG1..WFG9. k = num_objectives - 1 position parameters (matching
  the reference factory); dim - k must be even."""

  name: str = ''
  dim: int = 1
  num_objectives: int = 2

  def __post_init__(self):
    self.k = self.num_objectives - 1
    if (self.dim - self.k) % 2 != 0:
      raise ValueError(
          f'dimensions - k must be even, got {self.dim - self.k} for '
          f'k={self.k}.')
    if self.dim <= self.k:
      raise ValueError('WFG needs at least one distance parameter.')

  def __call__(self) -> MultiObjectiveNumpyExperimenter:
    impl = _wfg(self.name, self.num_objectives, self.dim, self.k)
    problem = _mo_problem_statement(self.dim, self.num_objectives)
    return MultiObjectiveNumpyExperimenter(impl, problem)

  def dump(self) -> vz.Metadata:
    md = vz.Metadata()
    md[EXPERIMENTER_FACTORY_KEY] = json.dumps({
        'name': self.name, 'dim': self.dim,
        'num_objectives': self.num_objectives})
    return md

  @classmethod
  def recover(cls, metadata: vz.Metadata) -> 'WFGExperimenterFactory':
    return cls(**json.loads(metadata[EXPERIMENTER_FACTORY_KEY]))
