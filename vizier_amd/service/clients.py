"""Public clients API (parity with vizier/service/clients/__init__.py)."""

from vizier_amd._src.service.clients import (
    ResourceNotFoundError,
    Study,
    Trial,
    UNUSED_CLIENT_ID,
    environment_variables,
)
from vizier_amd.client.client_abc import TrialIterable
