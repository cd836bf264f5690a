"""Service error taxonomy (parity with vizier/_src/service/custom_errors.py)."""


class CustomError(Exception):
  """Base class for service errors."""


class NotFoundError(CustomError, KeyError):
  """Raised when a resource does not exist in the datastore."""


class AlreadyExistsError(CustomError, KeyError):
  """Raised when creating a resource that already exists."""


class ImmutableStudyError(CustomError):
  """Raised on mutations to a non-ACTIVE study."""


class ImmutableTrialError(CustomError):
  """Raised on mutations to a completed/infeasible trial."""
