"""Public service API (parity with vizier/service in the reference)."""

from vizier_amd._src.service import clients
from vizier_amd._src.service import constants
from vizier_amd._src.service import resources
from vizier_amd._src.service.clients import Study, Trial, TrialIterable
from vizier_amd._src.service.policy_factory import DefaultPolicyFactory
from vizier_amd._src.service.pythia_service import PythiaServicer
from vizier_amd._src.service.vizier_client import (
    VizierClient,
    create_or_load_study,
    environment_variables,
)
from vizier_amd._src.service.vizier_server import (
    DefaultVizierServer,
    DistributedPythiaVizierServer,
)
from vizier_amd._src.service.vizier_service import VizierServicer

from vizier_amd._src.service.constants import (
    NO_ENDPOINT,
    SQL_LOCAL_URL,
    SQL_MEMORY_URL,
    VIZIER_DB_PATH,
)
