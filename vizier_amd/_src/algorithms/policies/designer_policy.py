"""Policy wrappers turning Designers into Pythia Policies.

Capability parity with vizier/_src/algorithms/policies/designer_policy.py
(DesignerPolicy :40, PartiallySerializableDesignerPolicy :126-377).
"""

from __future__ import annotations

import json
from typing import Callable, Optional

from vizier_amd._src.pyvizier.base_study_config import ProblemStatement
from vizier_amd._src.pyvizier.common import Metadata, Namespace
from vizier_amd._src.pyvizier.trial import MetadataDelta, TrialStatus
from vizier_amd._src.pythia import policy as pythia_policy
from vizier_amd._src.pythia.policy import (
    EarlyStopDecision,
    EarlyStopDecisions,
    EarlyStopRequest,
    SuggestDecision,
    SuggestRequest,
)
from vizier_amd._src.pythia.policy_supporter import PolicySupporter
from vizier_amd._src.algorithms.core.abstractions import (
    ActiveTrials,
    CompletedTrials,
    Designer,
    PartiallySerializableDesigner,
)

DesignerFactory = Callable[[ProblemStatement], Designer]


class DesignerPolicy(pythia_policy.Policy):
  """Stateless wrapper: rebuilds the designer on every suggest call."""

  def __init__(self, supporter: PolicySupporter,
               designer_factory: DesignerFactory):
    self._supporter = supporter
    self._designer_factory = designer_factory

  def suggest(self, request: SuggestRequest) -> SuggestDecision:
    designer = self._designer_factory(request.study_config)
    completed = self._supporter.GetTrials(
        study_guid=request.study_guid, status_matches=TrialStatus.COMPLETED)
    active = self._supporter.GetTrials(
        study_guid=request.study_guid, status_matches=TrialStatus.ACTIVE)
    designer.update(CompletedTrials(completed), ActiveTrials(active))
    suggestions = designer.suggest(request.count)
    return SuggestDecision(suggestions)


class _StateTracker:
  """Bookkeeping for incremental designer updates via study metadata."""

  NS = 'designer_policy_v0'

  def __init__(self, algorithm_ns: str):
    self._ns = (self.NS, algorithm_ns)

  def load(self, metadata: Metadata):
    view = metadata.abs_ns(self._ns)
    state = view.get('state', None)
    max_id = view.get('max_completed_id', None)
    return state, int(max_id) if max_id is not None else 0

  def updates(self, state_blob: str, max_completed_id: int) -> MetadataDelta:
    delta = MetadataDelta()
    view = delta.on_study.abs_ns(self._ns)
    view['state'] = state_blob
    view['max_completed_id'] = str(max_completed_id)
    return delta


class PartiallySerializableDesignerPolicy(pythia_policy.Policy):
  """Persists designer state in study metadata between suggest calls.

  The designer's `dump()` metadata is stored under a policy namespace; on
  the next call the designer is `load()`ed and fed only the trials
  completed since (incremental update). A failed load falls back to a
  full rebuild from all trials, matching designer_policy.py:266-312.
  """

  def __init__(self, problem: ProblemStatement, supporter: PolicySupporter,
               designer_factory: Callable[[ProblemStatement],
                                          PartiallySerializableDesigner],
               *, ns_root: str = 'designer'):
    self._problem = problem
    self._supporter = supporter
    self._designer_factory = designer_factory
    self._tracker = _StateTracker(ns_root)

  def suggest(self, request: SuggestRequest) -> SuggestDecision:
    designer = self._designer_factory(request.study_config)
    study_md = request.study_config.metadata
    state_blob, max_seen_id = self._tracker.load(study_md)

    incremental = False
    if state_blob is not None:
      try:
        designer.load(_metadata_from_blob(state_blob))
        incremental = True
      except Exception:
        incremental = False

    if incremental:
      completed = self._supporter.GetTrials(
          study_guid=request.study_guid, min_trial_id=max_seen_id + 1,
          status_matches=TrialStatus.COMPLETED)
    else:
      completed = self._supporter.GetTrials(
          study_guid=request.study_guid, status_matches=TrialStatus.COMPLETED)
    active = self._supporter.GetTrials(
        study_guid=request.study_guid, status_matches=TrialStatus.ACTIVE)

    designer.update(CompletedTrials(completed), ActiveTrials(active))
    suggestions = designer.suggest(request.count)

    new_max = max([t.id for t in completed], default=max_seen_id)
    delta = self._tracker.updates(_metadata_to_blob(designer.dump()), new_max)
    return SuggestDecision(suggestions, metadata=delta)


def _metadata_to_blob(md: Metadata) -> str:
  """Serializes a str-valued Metadata tree to JSON (proto values excluded)."""
  out = {}
  for ns in set(md.namespaces()) | {Namespace()}:
    view = md.abs_ns(ns)
    items = {k: v for k, v in view.items() if isinstance(v, str)}
    if items:
      out[ns.encode()] = items
  return json.dumps(out)


def _metadata_from_blob(blob: str) -> Metadata:
  md = Metadata()
  for ns_str, items in json.loads(blob).items():
    view = md.abs_ns(Namespace.decode(ns_str))
    for k, v in items.items():
      view[k] = v
  return md


class InRamDesignerPolicy(pythia_policy.Policy):
  """Keeps ONE designer instance in RAM and feeds it only new trials.

  Parity with designer_policy.py:347: no serialization — efficient for
  in-process services where the policy object survives across calls.
  """

  def __init__(self, problem: ProblemStatement, supporter: PolicySupporter,
               designer_factory: Callable[[ProblemStatement], object]):
    self._problem = problem
    self._supporter = supporter
    self._designer_factory = designer_factory
    self._designer = None
    self._max_seen_id = 0

  def suggest(self, request: SuggestRequest) -> SuggestDecision:
    if self._designer is None:
      self._designer = self._designer_factory(request.study_config)
      self._max_seen_id = 0
    completed = self._supporter.GetTrials(
        study_guid=request.study_guid, min_trial_id=self._max_seen_id + 1,
        status_matches=TrialStatus.COMPLETED)
    active = self._supporter.GetTrials(
        study_guid=request.study_guid, status_matches=TrialStatus.ACTIVE)
    self._designer.update(CompletedTrials(completed), ActiveTrials(active))
    self._max_seen_id = max([t.id for t in completed],
                            default=self._max_seen_id)
    return SuggestDecision(self._designer.suggest(request.count))


class SerializableDesignerPolicy(pythia_policy.Policy):
  """Wraps a fully SerializableDesigner: state restored via the class's
  recover() from study metadata, falling back to a fresh designer +
  full-history update (designer_policy.py:377)."""

  def __init__(self, problem: ProblemStatement, supporter: PolicySupporter,
               designer_factory: Callable[[ProblemStatement], object],
               designer_cls, *, ns_root: str = 'designer'):
    self._problem = problem
    self._supporter = supporter
    self._designer_factory = designer_factory
    self._designer_cls = designer_cls
    self._tracker = _StateTracker(ns_root)

  def suggest(self, request: SuggestRequest) -> SuggestDecision:
    study_md = request.study_config.metadata
    state_blob, max_seen_id = self._tracker.load(study_md)
    designer = None
    incremental = False
    if state_blob is not None:
      try:
        designer = self._designer_cls.recover(
            _metadata_from_blob(state_blob))
        incremental = True
      except Exception:
        designer = None
    if designer is None:
      designer = self._designer_factory(request.study_config)
    if incremental:
      completed = self._supporter.GetTrials(
          study_guid=request.study_guid, min_trial_id=max_seen_id + 1,
          status_matches=TrialStatus.COMPLETED)
    else:
      completed = self._supporter.GetTrials(
          study_guid=request.study_guid,
          status_matches=TrialStatus.COMPLETED)
    active = self._supporter.GetTrials(
        study_guid=request.study_guid, status_matches=TrialStatus.ACTIVE)
    designer.update(CompletedTrials(completed), ActiveTrials(active))
    suggestions = designer.suggest(request.count)
    new_max = max([t.id for t in completed], default=max_seen_id)
    delta = self._tracker.updates(_metadata_to_blob(designer.dump()),
                                  new_max)
    return SuggestDecision(suggestions, metadata=delta)

