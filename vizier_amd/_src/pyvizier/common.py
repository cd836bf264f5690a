"""Namespaced metadata shared by studies and trials.

MI355X-native rewrite of the reference semantics in
vizier/_src/pyvizier/shared/common.py (Namespace :90, Metadata :225).
The string encoding of namespaces is part of the wire contract (metadata
keys round-trip through the KeyValue proto), so `encode`/`decode` are
bit-compatible with the reference: ``Namespace(()).encode() == ''`` and
``Namespace(('a','b')).encode() == ':a:b'`` with ``':'`` escaped as ``'\\:'``.
"""

from __future__ import annotations

import re
from collections import abc
from typing import Dict, Iterable, Iterator, Optional, Tuple, Type, TypeVar, Union

from google.protobuf import any_pb2
from google.protobuf.message import Message

MetadataValue = Union[str, any_pb2.Any, Message]
_M = TypeVar('_M', bound=Message)

# Split an encoded namespace on colons that are NOT escaped by a backslash.
_SPLIT_RE = re.compile(r'(?<!\\):')


class Namespace(abc.Sequence):
  r"""An immutable tuple-of-strings path identifying a metadata namespace.

  ``Namespace.decode(s)`` parses an encoded string (':'-separated, '\\:'
  escapes a literal colon); ``encode()`` is its inverse. Constructing from an
  iterable treats components literally (no reserved characters).
  """

  __slots__ = ('_parts',)

  def __init__(self, arg: Iterable[str] = ()):
    if isinstance(arg, Namespace):
      object.__setattr__(self, '_parts', arg._parts)
    else:
      parts = tuple(arg)
      for p in parts:
        if not isinstance(p, str):
          raise TypeError(f'Namespace components must be str, got {p!r}')
      object.__setattr__(self, '_parts', parts)

  def __setattr__(self, *_):
    raise AttributeError('Namespace is immutable')

  def __reduce__(self):
    # Supports copy/deepcopy/pickle despite the immutability guard.
    return (Namespace, (self._parts,))

  @classmethod
  def decode(cls, s: str) -> 'Namespace':
    if not s:
      return cls(())
    if s.startswith(':'):
      s = s[1:]
    frags = _SPLIT_RE.split(s)
    return cls(f.replace('\\:', ':') for f in frags)

  def encode(self) -> str:
    return ''.join(':' + p.replace(':', '\\:') for p in self._parts)

  def __len__(self) -> int:
    return len(self._parts)

  def __getitem__(self, key):
    if isinstance(key, slice):
      return Namespace(self._parts[key])
    return self._parts[key]

  def __add__(self, other: Iterable[str]) -> 'Namespace':
    return Namespace(self._parts + tuple(other))

  def __eq__(self, other) -> bool:
    if isinstance(other, Namespace):
      return self._parts == other._parts
    return NotImplemented

  def __lt__(self, other: 'Namespace') -> bool:
    return self._parts < other._parts

  def __hash__(self) -> int:
    return hash(self._parts)

  def __str__(self) -> str:
    return self.encode()

  def __repr__(self) -> str:
    return f'Namespace({self.encode()})'

  def startswith(self, prefix: Iterable[str]) -> bool:
    p = Namespace(prefix)
    return self._parts[: len(p)] == tuple(p)


class Metadata(abc.MutableMapping):
  """A mapping str -> (str | proto) organized into a tree of namespaces.

  The object is a *view* onto one namespace of a shared store; ``ns(...)``
  returns a view of a sub-namespace backed by the same store, and
  ``abs_ns(...)`` returns a view of an absolute namespace.

  Mirrors vizier/_src/pyvizier/shared/common.py:225 capability-wise.
  """

  def __init__(self, *args, namespace: Iterable[str] = (), **kwargs):
    # Shared across all views of the same metadata tree.
    self._store: Dict[Namespace, Dict[str, MetadataValue]] = {}
    self._ns = Namespace(namespace)
    self._store.setdefault(self._ns, {})
    if args or kwargs:
      self._store[self._ns].update(dict(*args, **kwargs))

  # -- namespace navigation ------------------------------------------------

  @property
  def namespace(self) -> Namespace:
    return self._ns

  def ns(self, component: Union[str, Iterable[str]]) -> 'Metadata':
    """Returns a view of the sub-namespace `component` below the current one."""
    if isinstance(component, str):
      component = (component,)
    return self.abs_ns(self._ns + tuple(component))

  def abs_ns(self, namespace: Iterable[str] = ()) -> 'Metadata':
    """Returns a view of the absolute namespace `namespace`."""
    view = Metadata.__new__(Metadata)
    view._store = self._store
    view._ns = Namespace(namespace)
    view._store.setdefault(view._ns, {})
    return view

  def namespaces(self) -> Tuple[Namespace, ...]:
    """All non-empty absolute namespaces in the store."""
    return tuple(ns for ns, d in self._store.items() if d)

  def subnamespaces(self) -> Tuple[Namespace, ...]:
    """Non-empty namespaces that extend (or equal) the current namespace.

    Returned relative to the current namespace, like the reference.
    """
    out = []
    for ns, d in self._store.items():
      if d and ns.startswith(self._ns):
        out.append(Namespace(tuple(ns)[len(self._ns):]))
    return tuple(out)

  # -- mapping interface (current namespace only) --------------------------

  def _dict(self) -> Dict[str, MetadataValue]:
    return self._store.setdefault(self._ns, {})

  def __getitem__(self, key: str) -> MetadataValue:
    return self._dict()[key]

  def __setitem__(self, key: str, value: MetadataValue):
    if not isinstance(value, (str, Message)):
      raise TypeError(f'Metadata values must be str or proto; got {value!r}')
    self._dict()[key] = value

  def __delitem__(self, key: str):
    del self._dict()[key]

  def __iter__(self) -> Iterator[str]:
    return iter(self._dict())

  def __len__(self) -> int:
    return len(self._dict())

  def __bool__(self) -> bool:
    return any(bool(d) for d in self._store.values())

  def __eq__(self, other) -> bool:
    if not isinstance(other, Metadata):
      return NotImplemented
    mine = {ns: d for ns, d in self._store.items() if d}
    theirs = {ns: d for ns, d in other._store.items() if d}
    return self._ns == other._ns and mine == theirs

  # -- typed accessors -----------------------------------------------------

  def get_proto(self, key: str, *, cls: Type[_M]) -> Optional[_M]:
    """Gets the proto stored under `key`, unpacking Any if needed."""
    value = self._dict().get(key, None)
    if value is None:
      return None
    if isinstance(value, cls):
      return value
    if isinstance(value, any_pb2.Any):
      out = cls()
      if value.Unpack(out):
        return out
    return None

  def get(self, key: str, default=None, *, cls: Type = str):
    """Gets the value under `key` if it is an instance of `cls`."""
    if issubclass(cls, Message) and not issubclass(cls, any_pb2.Any):
      found = self.get_proto(key, cls=cls)
      return default if found is None else found
    value = self._dict().get(key, None)
    if isinstance(value, cls):
      return value
    return default

  def update(self, *args, **kwargs) -> None:
    self._dict().update(dict(*args, **kwargs))

  def attach(self, metadata: 'Metadata') -> None:
    """Merges another Metadata tree into this one (all namespaces)."""
    for ns, d in metadata._store.items():
      if d:
        self._store.setdefault(ns, {}).update(d)

  def __repr__(self) -> str:
    items = {str(ns): d for ns, d in self._store.items() if d}
    return f'Metadata(ns={self._ns.encode()!r}, store={items!r})'
