"""Pythia error taxonomy (parity with vizier/_src/pythia/pythia_errors.py)."""


class PythiaError(Exception):
  """Base class for Pythia errors."""


class TemporaryPythiaError(PythiaError):
  """Retryable failure inside a policy (e.g. numerical crash)."""


class InactivateStudyError(PythiaError):
  """The policy requests that the study be made INACTIVE."""


class CachedPolicyIsStaleError(PythiaError):
  """A cached policy is out of date and should be rebuilt."""


class LoadTooLargeError(PythiaError):
  """The request would use too much memory."""


class PythiaProtocolError(PythiaError):
  """An error in the Pythia protocol itself; not the user's fault."""


class VizierDatabaseError(PythiaError):
  """Vizier could not service a database request from a policy."""
