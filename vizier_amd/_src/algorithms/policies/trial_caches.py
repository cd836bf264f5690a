"""Incremental trial loading for designer policies.

Capability parity with vizier/_src/algorithms/policies/trial_caches.py
(IdDeduplicatingTrialLoader :33): wraps a PolicySupporter so repeated
Suggest() calls only fetch trials completed since the previous call,
with the incorporated-id set round-tripping through study metadata.
"""

from __future__ import annotations

import json
from typing import List, Set

from vizier_amd import pyvizier as vz
from vizier_amd.interfaces import serializable
from vizier_amd._src.pythia.policy_supporter import PolicySupporter

_INCORPORATED_COMPLETED_TRIAL_IDS = 'incorporated_completed_trials_ids'


class IdDeduplicatingTrialLoader(serializable.PartiallySerializable):
  """Returns each completed trial exactly once across calls."""

  def __init__(self, supporter: PolicySupporter, *,
               include_intermediate_measurements: bool = False):
    self._supporter = supporter
    self._include_intermediate = include_intermediate_measurements
    self._incorporated_ids: Set[int] = set()

  def num_incorporated_trials(self) -> int:
    return len(self._incorporated_ids)

  def clear(self) -> None:
    """Next get_newly_completed_trials() returns everything again."""
    self._incorporated_ids = set()

  def get_active_trials(self) -> List[vz.Trial]:
    return self._supporter.GetTrials(
        status_matches=vz.TrialStatus.ACTIVE,
        include_intermediate_measurements=self._include_intermediate)

  def get_newly_completed_trials(self, max_trial_id: int
                                 ) -> List[vz.Trial]:
    """Completed trials with id <= max_trial_id not yet returned."""
    if len(self._incorporated_ids) == max_trial_id:
      return []
    to_load = set(range(1, max_trial_id + 1)) - self._incorporated_ids
    if not to_load:
      return []
    new_trials = self._supporter.GetTrials(
        trial_ids=to_load,
        status_matches=vz.TrialStatus.COMPLETED,
        include_intermediate_measurements=self._include_intermediate)
    self._incorporated_ids |= {t.id for t in new_trials}
    return new_trials

  def dump(self) -> vz.Metadata:
    md = vz.Metadata()
    md[_INCORPORATED_COMPLETED_TRIAL_IDS] = json.dumps(
        sorted(self._incorporated_ids))
    return md

  def load(self, md: vz.Metadata) -> None:
    try:
      blob = md[_INCORPORATED_COMPLETED_TRIAL_IDS]
    except KeyError:
      self._incorporated_ids = set()
      return
    try:
      self._incorporated_ids = set(json.loads(blob))
    except (ValueError, TypeError) as e:
      raise serializable.HarmlessDecodeError(
          f'Corrupt trial-id cache: {e}') from e
