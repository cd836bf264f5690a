"""Service-wide constants (parity with vizier/_src/service/constants.py)."""

from __future__ import annotations

import os

# The metadata namespace under which the Pythia endpoint is stored.
PYTHIA_ENDPOINT_NAMESPACE = 'service'
# The Study.metadata key where a Pythia endpoint can be stored.
PYTHIA_ENDPOINT_KEY = 'PYTHIA_ENDPOINT'
# Indicates that the Pythia endpoint is not set (use in-process Pythia).
NO_ENDPOINT = 'NO_ENDPOINT'

UNUSED_CLIENT_ID = 'unused_client_id'
MAX_STUDY_ID = 2147483647

SQL_MEMORY_URL = 'sqlite:///:memory:'
SERVICE_DIR = os.path.dirname(os.path.realpath(__file__))
VIZIER_DB_PATH = os.path.join(SERVICE_DIR, 'vizier.db')
SQL_LOCAL_URL = f'sqlite:///{VIZIER_DB_PATH}'
