"""Helpers to move Metadata between pyvizier and the KeyValue wire format.

Capability parity with vizier/_src/pyvizier/oss/metadata_util.py.
"""

from __future__ import annotations

from typing import Dict, Iterable, List, Optional, Tuple, TypeVar, Union

from google.protobuf import any_pb2
from google.protobuf.message import Message

from vizier_amd._src.pyvizier.common import Metadata, Namespace
from vizier_amd._src.service.proto import study_pb2
from vizier_amd._src.service.proto import vizier_service_pb2

_KeyValue = study_pb2.KeyValue

T = TypeVar('T')


def _set_value(kv, value: Union[str, any_pb2.Any, Message]) -> None:
  if isinstance(value, str):
    kv.value = value
  elif isinstance(value, any_pb2.Any):
    kv.proto.CopyFrom(value)
  elif isinstance(value, Message):
    kv.proto.Pack(value)
  else:
    raise ValueError(f'Unsupported metadata value {value!r}')


def _get_value(kv) -> Union[str, any_pb2.Any]:
  if kv.HasField('proto'):
    return kv.proto
  return kv.value


def assign(container, *, key: str, ns: str, value,
           mode: str = 'insert_or_assign') -> Tuple[object, bool]:
  """Inserts or assigns (key, ns, value) in container.metadata.

  `container` is any proto with a repeated KeyValue `metadata` field.
  Returns (the kv entry, inserted: bool).
  """
  for kv in container.metadata:
    if kv.key == key and kv.ns == ns:
      if mode == 'insert':
        raise ValueError(f'Duplicate metadata key {key!r} in ns {ns!r}')
      _set_value(kv, value)
      return kv, False
  kv = container.metadata.add()
  kv.key = key
  kv.ns = ns
  _set_value(kv, value)
  return kv, True


def get(container, *, key: str, ns: str) -> Optional[Union[str, any_pb2.Any]]:
  for kv in container.metadata:
    if kv.key == key and kv.ns == ns:
      return _get_value(kv)
  return None


def to_key_value_protos(metadata: Metadata) -> List['study_pb2.KeyValue']:
  """Flattens a Metadata tree into KeyValue protos."""
  out = []
  # Include the root namespace explicitly (namespaces() skips empty dicts).
  for ns in set(metadata.namespaces()) | {Namespace()}:
    view = metadata.abs_ns(ns)
    for key, value in view.items():
      kv = _KeyValue(key=key, ns=ns.encode())
      _set_value(kv, value)
      out.append(kv)
  return out


def from_key_value_protos(kvs: Iterable['study_pb2.KeyValue']) -> Metadata:
  md = Metadata()
  for kv in kvs:
    md.abs_ns(Namespace.decode(kv.ns))[kv.key] = _get_value(kv)
  return md


def study_metadata_to_update_list(
    metadata: Metadata) -> List['vizier_service_pb2.UnitMetadataUpdate']:
  out = []
  for kv in to_key_value_protos(metadata):
    u = vizier_service_pb2.UnitMetadataUpdate()
    u.metadatum.CopyFrom(kv)
    out.append(u)
  return out


def trial_metadata_to_update_list(
    trial_metadata: Dict[int, Metadata]
) -> List['vizier_service_pb2.UnitMetadataUpdate']:
  out = []
  for trial_id, metadata in trial_metadata.items():
    for kv in to_key_value_protos(metadata):
      u = vizier_service_pb2.UnitMetadataUpdate(trial_id=str(trial_id))
      u.metadatum.CopyFrom(kv)
      out.append(u)
  return out


def merge_study_metadata(study_spec, new_metadata:
                         Iterable['study_pb2.KeyValue']) -> None:
  """Insert-or-updates metadata into the study spec proto."""
  for kv in new_metadata:
    assign(study_spec, key=kv.key, ns=kv.ns, value=_get_value(kv))


def merge_trial_metadata(trial_proto, new_metadata:
                         Iterable['study_pb2.KeyValue']) -> None:
  for kv in new_metadata:
    assign(trial_proto, key=kv.key, ns=kv.ns, value=_get_value(kv))
