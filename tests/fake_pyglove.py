"""Minimal in-repo stand-in for the `pyglove` package.

Implements exactly the pg API subset the vizier_amd pyglove plugin
consumes (documented in vizier_amd/_src/pyglove/converters.py), so the
plugin's code paths — DNA spec conversion, spec persistence, the
TunerPolicy loop, the backend's chief election — are all testable in
this offline image. Install with `install()` BEFORE importing the
plugin modules.
"""

from __future__ import annotations

import json
import sys
import types
from typing import Any, Dict, List, Optional


# -- geno specs ---------------------------------------------------------------


class _SpecBase:

  def to_json(self):
    raise NotImplementedError


class Float(_SpecBase):

  def __init__(self, min_value, max_value, *, name=None, scale=None):
    self.min_value = float(min_value)
    self.max_value = float(max_value)
    self.name = name
    self.location = name
    self.scale = scale

  def to_json(self):
    return {'_type': 'float', 'min': self.min_value,
            'max': self.max_value, 'name': self.name,
            'scale': self.scale}


class Choices(_SpecBase):

  def __init__(self, num_choices, candidates, *, literal_values=None,
               name=None):
    self.num_choices = num_choices
    self.candidates = list(candidates)
    self.literal_values = list(literal_values or [])
    self.name = name
    self.location = name

  def format_candidate(self, i):
    return str(self.literal_values[i])

  def to_json(self):
    return {'_type': 'choices', 'num': self.num_choices,
            'candidates': [c.to_json() for c in self.candidates],
            'literal_values': self.literal_values, 'name': self.name}


class Space(_SpecBase):

  def __init__(self, elements=()):
    self.elements = list(elements)
    self.name = None
    self.location = None
    self.hints = None

  def to_json(self):
    return {'_type': 'space',
            'elements': [e.to_json() for e in self.elements],
            'hints': self.hints}


class CustomDecisionPoint(_SpecBase):

  def __init__(self, name=None):
    self.name = name
    self.location = name

  def to_json(self):
    return {'_type': 'custom', 'name': self.name}


def _spec_from_json(d):
  t = d['_type']
  if t == 'float':
    return Float(d['min'], d['max'], name=d['name'], scale=d['scale'])
  if t == 'choices':
    return Choices(d['num'], [_spec_from_json(c) for c in d['candidates']],
                   literal_values=d['literal_values'], name=d['name'])
  if t == 'space':
    s = Space([_spec_from_json(e) for e in d['elements']])
    s.hints = d.get('hints')
    return s
  if t == 'custom':
    return CustomDecisionPoint(name=d['name'])
  raise ValueError(t)


# -- DNA ----------------------------------------------------------------------


class DNA:
  """Named-decision DNA: a dict of decision-point name -> value."""

  def __init__(self, decisions: Optional[Dict[str, Any]] = None,
               metadata: Optional[Dict[str, Any]] = None):
    self.decisions = dict(decisions or {})
    self.metadata = dict(metadata or {})
    self.spec = None

  @classmethod
  def from_dict(cls, d, spec, use_ints_as_literals=False):
    dna = cls(dict(d))
    dna.spec = spec
    return dna

  def to_dict(self, key_type='name_or_id', value_type='literal'):
    return dict(self.decisions)

  def use_spec(self, spec):
    self.spec = spec
    return self

  def rebind(self, metadata=None, **kwargs):
    if metadata is not None:
      self.metadata = dict(metadata)

  def __str__(self):
    return f'DNA({self.decisions})'

  def __eq__(self, other):
    return isinstance(other, DNA) and other.decisions == self.decisions


class DNAGenerator:
  """pg.DNAGenerator protocol subset."""

  def setup(self, dna_spec):
    self._dna_spec = dna_spec

  @property
  def dna_spec(self):
    return getattr(self, '_dna_spec', None)

  def propose(self):
    return self._propose()

  def _propose(self):
    raise NotImplementedError

  def feedback(self, dna, reward):
    self._feedback(dna, reward)

  def _feedback(self, dna, reward):
    pass

  def recover(self, history):
    for dna, reward in history:
      if reward is not None:
        self.feedback(dna, reward)

  def __eq__(self, other):
    # pyglove symbolic equality compares by value; generators of the
    # same class with the same (symbolic) config are equal. The shim
    # compares by class, which is what the backend's same-study
    # algorithm check needs.
    return type(other) is type(self)

  def __hash__(self):
    return hash(type(self))


# -- json ---------------------------------------------------------------------


def to_json(x):
  if isinstance(x, _SpecBase):
    return x.to_json()
  return x


def from_json(d):
  if isinstance(d, dict) and '_type' in d:
    return _spec_from_json(d)
  return d


def to_json_str(x):
  return json.dumps(to_json(x))


def from_json_str(s):
  return from_json(json.loads(s))


# -- tuning namespace ---------------------------------------------------------


class _TuningNS(types.SimpleNamespace):

  def __init__(self):
    super().__init__()
    self.backends = {}
    self.default_backend = None

    ns = self

    class EarlyStoppingPolicy:
      def should_stop_early(self, trial) -> bool:
        return False

    def add_backend(name):
      def register(cls):
        ns.backends[name] = cls
        return cls
      return register

    def set_default_backend(name):
      ns.default_backend = name

    self.EarlyStoppingPolicy = EarlyStoppingPolicy
    self.add_backend = add_backend
    self.set_default_backend = set_default_backend


def install():
  """Installs this shim as `pyglove` in sys.modules (idempotent)."""
  if 'pyglove' in sys.modules and getattr(
      sys.modules['pyglove'], '_IS_VIZIER_AMD_FAKE', False):
    return sys.modules['pyglove']
  mod = types.ModuleType('pyglove')
  mod._IS_VIZIER_AMD_FAKE = True
  geno = types.ModuleType('pyglove.geno')
  geno.Float = Float
  geno.Choices = Choices
  geno.Space = Space
  geno.CustomDecisionPoint = CustomDecisionPoint
  mod.geno = geno
  mod.DNA = DNA
  mod.DNASpec = _SpecBase
  mod.DNAGenerator = DNAGenerator
  mod.to_json = to_json
  mod.from_json = from_json
  mod.to_json_str = to_json_str
  mod.from_json_str = from_json_str
  mod.tuning = _TuningNS()
  sys.modules['pyglove'] = mod
  sys.modules['pyglove.geno'] = geno
  return mod
