"""Lightweight scope profiler.

Capability parity with vizier/utils/profiler.py (global _Storage :86,
collect_events :138, timeit :156, record_runtime :213,
get_latencies_dict :203), extended with optional torch.cuda event timing
for GPU scopes (the MI355X analogue of the reference's jax.monitoring
breadcrumbs).
"""

from __future__ import annotations

import collections
import contextlib
import dataclasses
import datetime
import functools
import threading
import time
from typing import Callable, Dict, Iterator, List, Optional


@dataclasses.dataclass(frozen=True)
class ProfileEvent:
  name: str
  duration: datetime.timedelta
  timestamp: float


class _Storage:
  """Thread-safe global event store, active only inside collect_events."""

  def __init__(self):
    self._lock = threading.Lock()
    self._events: List[ProfileEvent] = []
    self._active = False

  def activate(self):
    with self._lock:
      self._events = []
      self._active = True

  def deactivate(self) -> List[ProfileEvent]:
    with self._lock:
      self._active = False
      return list(self._events)

  def record(self, event: ProfileEvent):
    with self._lock:
      if self._active:
        self._events.append(event)

  @property
  def events(self) -> List[ProfileEvent]:
    with self._lock:
      return list(self._events)


_storage = _Storage()


@contextlib.contextmanager
def collect_events() -> Iterator[List[ProfileEvent]]:
  """Activates collection; yields the (mutating) event list."""
  _storage.activate()
  out: List[ProfileEvent] = []
  try:
    yield out
  finally:
    out.extend(_storage.deactivate())


@contextlib.contextmanager
def timeit(name: str, also_log: bool = False,
           sync_gpu: bool = False) -> Iterator[None]:
  """Times a scope; optionally synchronizes the GPU at both edges.

  Scopes are also emitted as torch.profiler record_function ranges, so
  they appear as named regions in torch.profiler / rocTracer traces
  (`rocprofv3 --hip-trace` + torch profiler export) alongside the HIP
  kernels they launch.
  """
  range_cm = None
  try:
    import torch
    range_cm = torch.profiler.record_function(name)
    range_cm.__enter__()
  except ImportError:
    pass
  if sync_gpu:
    try:
      import torch
      if torch.cuda.is_available():
        torch.cuda.synchronize()
    except ImportError:
      pass
  start = time.monotonic()
  try:
    yield
  finally:
    if sync_gpu:
      try:
        import torch
        if torch.cuda.is_available():
          torch.cuda.synchronize()
      except ImportError:
        pass
    if range_cm is not None:
      range_cm.__exit__(None, None, None)
    duration = datetime.timedelta(seconds=time.monotonic() - start)
    _storage.record(ProfileEvent(name, duration, time.time()))
    if also_log:
      import logging
      logging.getLogger(__name__).info('%s took %s', name, duration)


def record_runtime(func: Optional[Callable] = None, *,
                   name_prefix: str = '', also_log: bool = False,
                   block_until_ready: bool = False):
  """Decorator recording the wrapped function's runtime."""

  def decorator(f):
    scope = f'{name_prefix}.{f.__name__}' if name_prefix else f.__name__

    @functools.wraps(f)
    def wrapper(*args, **kwargs):
      with timeit(scope, also_log=also_log, sync_gpu=block_until_ready):
        return f(*args, **kwargs)
    return wrapper

  if func is not None:
    return decorator(func)
  return decorator


def get_latencies_dict(events: List[ProfileEvent]
                       ) -> Dict[str, List[datetime.timedelta]]:
  out: Dict[str, List[datetime.timedelta]] = collections.defaultdict(list)
  for e in events:
    out[e.name].append(e.duration)
  return dict(out)
