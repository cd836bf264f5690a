"""Public servers API (parity with vizier/service/servers/__init__.py)."""

from vizier_amd._src.service.vizier_server import (
    DefaultVizierServer,
    DistributedPythiaVizierServer,
)
