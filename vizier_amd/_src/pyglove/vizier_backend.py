"""PyGlove integration: evolutionary backends on this Vizier service.

Capability parity (in scope) with vizier/_src/pyglove/
(OSSVizierBackend oss_vizier.py:290, TunerPolicy pythia.py:33,
DNA<->Trial converters converters.py, chief election backend.py:410-466).
`pyglove` is not installed in this image, so everything that touches it
is import-deferred; the DNA encoding used here is pyglove's JSON-compat
`to_json_str` formatted DNA stored in trial metadata.
"""

from __future__ import annotations

from typing import Any, Dict, Optional, Sequence

from vizier_amd import pyvizier as vz
from vizier_amd.service import clients

_NS = 'pyglove'
_DNA_KEY = 'dna'
_CHIEF_KEY = 'chief_id'


def _require_pyglove():
  try:
    import pyglove as pg  # Deferred optional dependency.
    return pg
  except ImportError as e:
    raise ImportError(
        'The pyglove integration requires the `pyglove` package.') from e


class VizierTuner:
  """Feeds a pyglove DNAGenerator through a Vizier study.

  Each DNA proposal becomes a REQUESTED trial carrying its serialized
  DNA in metadata; rewards flow back to the generator on completion.
  Multiple worker processes sharing (owner, study_id) elect a chief via
  study metadata: the first writer of `chief_id` wins (parity with
  backend.py:410-466).
  """

  def __init__(self, study: clients.Study, worker_id: str):
    self._study = study
    self._worker_id = worker_id

  def elect_chief(self) -> bool:
    """Returns True if this worker becomes (or already is) the chief."""
    config = self._study.materialize_study_config()
    current = config.metadata.ns(_NS).get(_CHIEF_KEY, None)
    if current is None:
      md = vz.Metadata()
      md.ns(_NS)[_CHIEF_KEY] = self._worker_id
      self._study.update_metadata(md)
      config = self._study.materialize_study_config()
      current = config.metadata.ns(_NS).get(_CHIEF_KEY, None)
    return current == self._worker_id

  def propose(self, generator, uid_hint: int = 0) -> 'clients.Trial':
    """Asks the DNA generator for a proposal and registers it."""
    pg = _require_pyglove()
    dna = generator.propose()
    suggestion = vz.TrialSuggestion(
        dna_to_parameters(dna))
    suggestion.metadata.ns(_NS)[_DNA_KEY] = pg.to_json_str(dna)
    return self._study.request(suggestion)

  def feedback(self, generator, trial: vz.Trial, metric_name: str
               ) -> None:
    pg = _require_pyglove()
    blob = trial.metadata.ns(_NS).get(_DNA_KEY, None)
    if blob is None or trial.final_measurement is None:
      return
    dna = pg.from_json_str(blob)
    reward = trial.final_measurement.metrics[metric_name].value
    generator.feedback(dna, reward)


def dna_to_parameters(dna) -> Dict[str, Any]:
  """Flattens a pyglove DNA into Vizier trial parameters."""
  params: Dict[str, Any] = {}
  for i, value in enumerate(_dna_values(dna)):
    params[f'dna_{i}'] = value
  return params


def _dna_values(dna):
  out = []

  def walk(node):
    if node.value is not None:
      out.append(node.value)
    for child in node.children:
      walk(child)

  walk(dna)
  return out
