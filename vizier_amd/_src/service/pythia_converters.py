"""Converters between Pythia dataclasses and the pythia_service protos.

Capability parity with the SuggestConverter/EarlyStopConverter pieces of
vizier/_src/pyvizier/oss/proto_converters.py:869,927.
"""

from __future__ import annotations

from typing import Dict, List

from vizier_amd._src.pyvizier import metadata_util
from vizier_amd._src.pyvizier import proto_converters as pc
from vizier_amd._src.pyvizier.common import Metadata, Namespace
from vizier_amd._src.pyvizier.study_config import StudyConfig
from vizier_amd._src.pyvizier.trial import MetadataDelta
from vizier_amd._src.pythia import policy as pythia
from vizier_amd._src.service.proto import (
    pythia_service_pb2,
    study_pb2,
    vizier_service_pb2,
)


def _metadata_delta_to_updates(delta: MetadataDelta) -> List:
  out = list(metadata_util.study_metadata_to_update_list(delta.on_study))
  out.extend(metadata_util.trial_metadata_to_update_list(delta.on_trials))
  return out


def _updates_to_metadata_delta(updates) -> MetadataDelta:
  delta = MetadataDelta()
  for u in updates:
    kv = u.metadatum
    value = kv.proto if kv.HasField('proto') else kv.value
    if u.HasField('trial_id'):
      md = delta.on_trials.setdefault(int(u.trial_id), Metadata())
      md.abs_ns(Namespace.decode(kv.ns))[kv.key] = value
    else:
      delta.on_study.abs_ns(Namespace.decode(kv.ns))[kv.key] = value
  return delta


def _study_descriptor_to_proto(d: pythia.StudyDescriptor):
  proto = pythia_service_pb2.StudyDescriptor(
      guid=d.guid, max_trial_id=d.max_trial_id)
  proto.config.CopyFrom(pc.ProblemStatementConverter.to_proto(
      d.config.to_problem() if hasattr(d.config, 'to_problem') else d.config))
  return proto


def _study_descriptor_from_proto(proto, algorithm: str
                                 ) -> pythia.StudyDescriptor:
  problem = pc.ProblemStatementConverter.from_proto(proto.config)
  config = StudyConfig.from_problem(problem)
  config.algorithm = algorithm
  return pythia.StudyDescriptor(config=config, guid=proto.guid,
                                max_trial_id=proto.max_trial_id)


class SuggestConverter:
  """pythia.SuggestRequest/Decision <-> protos."""

  @classmethod
  def to_request_proto(cls, request: pythia.SuggestRequest):
    proto = pythia_service_pb2.SuggestRequest(count=request.count)
    proto.study_descriptor.CopyFrom(
        _study_descriptor_to_proto(request.study_descriptor))
    if request.checkpoint_dir:
      proto.checkpoint_dir = request.checkpoint_dir
    return proto

  @classmethod
  def from_request_proto(cls, proto) -> pythia.SuggestRequest:
    return pythia.SuggestRequest(
        study_descriptor=_study_descriptor_from_proto(proto.study_descriptor,
                                                      proto.algorithm),
        count=proto.count,
        checkpoint_dir=proto.checkpoint_dir or None)

  @classmethod
  def to_decision_proto(cls, decision: pythia.SuggestDecision):
    proto = pythia_service_pb2.SuggestDecision()
    for s in decision.suggestions:
      proto.suggestions.add().CopyFrom(pc.TrialSuggestionConverter.to_proto(s))
    for u in _metadata_delta_to_updates(decision.metadata):
      proto.metadata.add().CopyFrom(u)
    return proto

  @classmethod
  def from_decision_proto(cls, proto) -> pythia.SuggestDecision:
    suggestions = [pc.TrialSuggestionConverter.from_proto(s)
                   for s in proto.suggestions]
    return pythia.SuggestDecision(
        suggestions, metadata=_updates_to_metadata_delta(proto.metadata))


class EarlyStopConverter:
  """pythia.EarlyStopRequest/Decisions <-> protos."""

  @classmethod
  def to_request_proto(cls, request: pythia.EarlyStopRequest):
    proto = pythia_service_pb2.EarlyStopRequest(
        trial_ids=sorted(request.trial_ids))
    proto.study_descriptor.CopyFrom(
        _study_descriptor_to_proto(request.study_descriptor))
    if request.checkpoint_dir:
      proto.checkpoint_dir = request.checkpoint_dir
    return proto

  @classmethod
  def from_request_proto(cls, proto) -> pythia.EarlyStopRequest:
    return pythia.EarlyStopRequest(
        study_descriptor=_study_descriptor_from_proto(proto.study_descriptor,
                                                      proto.algorithm),
        trial_ids=frozenset(proto.trial_ids),
        checkpoint_dir=proto.checkpoint_dir or None)

  @classmethod
  def to_decisions_proto(cls, decisions: pythia.EarlyStopDecisions):
    proto = pythia_service_pb2.EarlyStopDecisions()
    for d in decisions.decisions:
      dp = proto.decisions.add(id=d.id, reason=d.reason,
                               should_stop=d.should_stop)
      if d.predicted_final_measurement is not None:
        dp.predicted_final_measurement.CopyFrom(
            pc.MeasurementConverter.to_proto(d.predicted_final_measurement))
    for u in _metadata_delta_to_updates(decisions.metadata):
      proto.metadata.add().CopyFrom(u)
    return proto

  @classmethod
  def from_decisions_proto(cls, proto) -> pythia.EarlyStopDecisions:
    decisions = []
    for dp in proto.decisions:
      pfm = None
      if dp.HasField('predicted_final_measurement'):
        pfm = pc.MeasurementConverter.from_proto(
            dp.predicted_final_measurement)
      decisions.append(pythia.EarlyStopDecision(
          id=dp.id, reason=dp.reason, should_stop=dp.should_stop,
          predicted_final_measurement=pfm))
    return pythia.EarlyStopDecisions(
        decisions, metadata=_updates_to_metadata_delta(proto.metadata))
