"""PyGlove DNA <-> Vizier Trial converters.

Capability parity with vizier/_src/pyglove/converters.py
(VizierConverter :254, _make_decision_point :56, _to_search_space :106):
bidirectional mapping between `pg.DNASpec` (geno.Float / geno.Choices /
geno.Space trees) and vizier SearchSpace/Trial, DNA-spec persistence in
study metadata (same compressed-JSON encoding + keys as the reference,
so studies interoperate), and the maximize-only metric contract.

Requires `pyglove` at import (the integration is dependency-gated like
the reference's); tests exercise it through a minimal pg shim.
"""

from __future__ import annotations

import base64
import json
import lzma
import numbers
from typing import Any, Dict, List, Optional, Sequence

import pyglove as pg

from vizier_amd import pyvizier as vz
from vizier_amd._src.pyglove import constants


def to_json_str_compressed(value: Any) -> str:
  """lzma+b64 of pg.to_json — the reference's on-wire spec encoding."""
  return base64.b64encode(
      lzma.compress(json.dumps(pg.to_json(value)).encode('utf-8'))
  ).decode('ascii')


def restore_dna_spec(json_str_compressed: str) -> 'pg.DNASpec':
  return pg.from_json(json.loads(
      lzma.decompress(base64.b64decode(json_str_compressed)).decode(
          'utf-8')))


def get_scale_type(scale: Optional[str]) -> Optional[vz.ScaleType]:
  """pg scale string -> vizier ScaleType (converters.py:214)."""
  if scale in (None, 'linear'):
    return vz.ScaleType.LINEAR
  if scale == 'log':
    return vz.ScaleType.LOG
  if scale == 'rlog':
    return vz.ScaleType.REVERSE_LOG
  raise ValueError(f'Unsupported scale type: {scale!r}')


def _scale_string(scale_type: Optional[vz.ScaleType]) -> Optional[str]:
  if scale_type in (None, vz.ScaleType.LINEAR):
    return None
  if scale_type == vz.ScaleType.LOG:
    return 'log'
  if scale_type == vz.ScaleType.REVERSE_LOG:
    return 'rlog'
  return None


def _make_decision_point(pc: vz.ParameterConfig):
  """ParameterConfig -> pg decision point (converters.py:56-100)."""
  name = pc.name
  if pc.type == vz.ParameterType.DOUBLE:
    lo, hi = pc.bounds
    return pg.geno.Float(lo, hi, name=name,
                         scale=_scale_string(pc.scale_type))
  if pc.type in (vz.ParameterType.CATEGORICAL,
                 vz.ParameterType.DISCRETE, vz.ParameterType.INTEGER):
    candidates = []
    literal_values = []
    if pc.type == vz.ParameterType.INTEGER:
      lo, hi = pc.bounds
      values = list(range(int(lo), int(hi) + 1))
    else:
      values = list(pc.feasible_values)
    for val in values:
      children = []
      for child in pc.child_parameter_configs:
        if val in child.matching_parent_values:
          children.append(_make_decision_point(child))
      candidates.append(pg.geno.Space(children))
      if pc.type == vz.ParameterType.INTEGER:
        literal_values.append(int(val))
      elif pc.type == vz.ParameterType.DISCRETE:
        literal_values.append(float(val))
      else:
        literal_values.append(val)
    return pg.geno.Choices(1, candidates,
                           literal_values=literal_values, name=name)
  raise ValueError(f'Parameter type {pc.type!r} is not supported.')


def to_dna_spec(search_space: vz.SearchSpace) -> 'pg.DNASpec':
  return pg.geno.Space(
      [_make_decision_point(pc) for pc in search_space.parameters])


def to_search_space(dna_spec) -> vz.SearchSpace:
  """pg.DNASpec -> vizier SearchSpace (converters.py:106-209).

  Decision points must be named (the reference names them too when it
  creates them; externally-built specs without names raise).
  """
  search_space = vz.SearchSpace()

  def _name_of(spec, path: str) -> str:
    name = getattr(spec, 'name', None) or path
    return name or constants.PARAMETER_NAME_ROOT

  def _add(root, path: str, spec) -> None:
    if isinstance(spec, pg.geno.Space):
      for elem in getattr(spec, 'elements', []):
        loc = getattr(elem, 'location', None) or getattr(elem, 'name', '')
        sub = f'{path}.{loc}' if path and loc else (loc or path)
        _add(root, sub, elem)
      return
    if isinstance(spec, pg.geno.Choices):
      literals = list(spec.literal_values)
      is_discrete = (all(isinstance(v, numbers.Number) for v in literals)
                     and len(set(literals)) == len(literals))
      if is_discrete:
        root.add_discrete_param(
            name=_name_of(spec, path),
            feasible_values=sorted(set(float(v) for v in literals)))
      else:
        selector = root.add_categorical_param(
            name=_name_of(spec, path),
            feasible_values=[str(v) for v in literals])
        for idx, candidate in enumerate(spec.candidates):
          if getattr(candidate, 'elements', None):
            child = selector.select_values([str(literals[idx])])
            _add(child, f'{_name_of(spec, path)}={literals[idx]}',
                 candidate)
      return
    if isinstance(spec, pg.geno.Float):
      root.add_float_param(
          name=_name_of(spec, path),
          min_value=spec.min_value, max_value=spec.max_value,
          scale_type=get_scale_type(getattr(spec, 'scale', None)))
      return
    if isinstance(spec, getattr(pg.geno, 'CustomDecisionPoint', ())):
      return  # value carried in metadata, not a Vizier parameter
    raise NotImplementedError(f'Unknown spec node: {spec!r}')

  _add(search_space.root, '', dna_spec)
  if not search_space.parameters:
    raise NotImplementedError(
        'No part of the dna spec could be represented as a Vizier '
        'parameter.')
  return search_space


def _iter_named_decisions(dna_spec) -> List:
  """Flattens named decision points of a spec tree."""
  out = []

  def _walk(spec):
    if isinstance(spec, pg.geno.Space):
      for e in getattr(spec, 'elements', []):
        _walk(e)
    else:
      out.append(spec)
      if isinstance(spec, pg.geno.Choices):
        for cand in spec.candidates:
          _walk(cand)

  _walk(dna_spec)
  return out


def get_pyglove_metadata(trial: vz.Trial) -> Dict[str, Any]:
  """Trial metadata -> plain dict (converters.py:226)."""
  metadata = {}
  for key, value in trial.metadata.items():
    if key in constants.TRIAL_METADATA_KEYS and value is not None:
      metadata[key] = pg.from_json_str(value)
  for key, value in trial.metadata.ns(
      constants.METADATA_NAMESPACE).items():
    if key not in constants.TRIAL_METADATA_KEYS and value is not None:
      metadata[key] = pg.from_json_str(value)
  return metadata


class VizierConverter:
  """Converts between pg.DNA and vz.Trial (converters.py:254).

  Use the factories `from_problem` / `from_dna_spec`. When a DNA spec
  has no Vizier representation, `vizier_conversion_error` is set and
  the problem contains the dummy parameter (Vizier algorithms cannot
  run, pyglove algorithms still can).
  """

  def __init__(self, dna_spec, problem: vz.ProblemStatement,
               uses_external_dna_spec: bool, *,
               vizier_conversion_error: Optional[Exception] = None):
    self._dna_spec = dna_spec
    self._problem = problem
    self._uses_external_dna_spec = uses_external_dna_spec
    self.vizier_conversion_error = vizier_conversion_error
    ns = self._problem.metadata.ns(constants.METADATA_NAMESPACE)
    ns[constants.STUDY_METADATA_KEY_DNA_SPEC] = \
        to_json_str_compressed(dna_spec)
    ns[constants.STUDY_METADATA_KEY_USES_EXTERNAL_DNA_SPEC] = \
        pg.to_json_str(uses_external_dna_spec)

  # -- factories ------------------------------------------------------------

  @classmethod
  def from_problem(cls, problem: vz.ProblemStatement) -> 'VizierConverter':
    ns = problem.metadata.ns(constants.METADATA_NAMESPACE)
    stored = ns.get(constants.STUDY_METADATA_KEY_DNA_SPEC, None)
    if stored is not None:
      dna_spec = restore_dna_spec(stored)
      uses_external = True
    else:
      dna_spec = to_dna_spec(problem.search_space)
      dna_spec.hints = constants.FROM_VIZIER_STUDY_HINT
      uses_external = False
    bad = [m for m in problem.metric_information
           if m.goal != vz.ObjectiveMetricGoal.MAXIMIZE]
    if bad:
      raise ValueError(f'All goals must MAXIMIZE. Offending: {bad}')
    return cls(dna_spec, problem, uses_external)

  @classmethod
  def from_dna_spec(cls, dna_spec, metrics_to_maximize: Sequence[str] = (
      constants.REWARD_METRIC_NAME,)) -> 'VizierConverter':
    problem = vz.ProblemStatement()
    for name in metrics_to_maximize:
      problem.metric_information.append(vz.MetricInformation(
          name=name, goal=vz.ObjectiveMetricGoal.MAXIMIZE))
    try:
      problem.search_space = to_search_space(dna_spec)
      return cls(dna_spec, problem, True)
    except NotImplementedError as e:
      problem.search_space.root.add_categorical_param(
          constants.DUMMY_PARAMETER_NAME,
          feasible_values=[constants.DUMMY_PARAMETER_VALUE])
      return cls(dna_spec, problem, True, vizier_conversion_error=e)

  # -- properties -----------------------------------------------------------

  @property
  def dna_spec(self):
    return self._dna_spec

  @property
  def problem(self) -> vz.ProblemStatement:
    if self.vizier_conversion_error:
      raise self.vizier_conversion_error
    return self._problem

  @property
  def problem_or_dummy(self) -> vz.ProblemStatement:
    return self._problem

  @property
  def search_space(self) -> vz.SearchSpace:
    return self.problem.search_space

  @property
  def uses_external_dna_spec(self) -> bool:
    return self._uses_external_dna_spec

  @property
  def metrics_to_optimize(self) -> List[str]:
    out = []
    for m in self._problem.metric_information:
      out.append(m.name if m.goal == vz.ObjectiveMetricGoal.MAXIMIZE
                 else f'negative_{m.name}')
    return out

  # -- DNA <-> Trial ---------------------------------------------------------

  def to_dna(self, trial: vz.Trial) -> 'pg.DNA':
    """Vizier trial -> DNA via the named-decision dict."""
    decisions = dict(trial.parameters.as_dict())
    decisions.pop(constants.DUMMY_PARAMETER_NAME, None)
    if constants.PARAMETER_NAME_ROOT in decisions:
      decisions[''] = decisions.pop(constants.PARAMETER_NAME_ROOT)
    custom = trial.metadata.ns(constants.METADATA_NAMESPACE).get(
        constants.TRIAL_METADATA_KEY_CUSTOM_TYPE_DECISIONS, None)
    if custom is not None:
      decisions.update(pg.from_json_str(custom))
    dna = pg.DNA.from_dict(decisions, self._dna_spec,
                           use_ints_as_literals=True)
    dna_metadata = trial.metadata.ns(constants.METADATA_NAMESPACE).get(
        constants.TRIAL_METADATA_KEY_DNA_METADATA, None)
    if dna_metadata is None:
      dna_metadata = trial.metadata.get(
          constants.TRIAL_METADATA_KEY_DNA_METADATA, None)
    if dna_metadata is not None:
      try:
        dna.rebind(metadata=pg.from_json_str(dna_metadata),
                   skip_notification=True, raise_on_no_change=False)
      except (AttributeError, TypeError):
        dna.metadata = pg.from_json_str(dna_metadata)
    return dna

  def to_trial(self, dna, *, fallback: str = 'raise_error') -> vz.Trial:
    """DNA -> vizier trial (converters.py:444-520)."""
    trial = vz.Trial()
    trial.description = str(dna)
    trial.metadata.ns(constants.METADATA_NAMESPACE)[
        constants.TRIAL_METADATA_KEY_DNA_METADATA] = pg.to_json_str(
            dict(getattr(dna, 'metadata', {}) or {}))
    if self.vizier_conversion_error:
      if fallback == 'raise_error':
        raise self.vizier_conversion_error
      trial.parameters[constants.DUMMY_PARAMETER_NAME] = \
          constants.DUMMY_PARAMETER_VALUE
      return trial
    if getattr(dna, 'spec', None) is None and hasattr(dna, 'use_spec'):
      dna.use_spec(self._dna_spec)
    try:
      decisions = dna.to_dict(key_type='name_or_id',
                              value_type='literal')
    except TypeError:
      decisions = dna.to_dict()
    for key, value in decisions.items():
      name = key or constants.PARAMETER_NAME_ROOT
      trial.parameters[name] = value
    return trial
