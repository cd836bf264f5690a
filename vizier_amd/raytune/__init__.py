"""Public Ray Tune integration (parity with vizier/raytune)."""

from vizier_amd._src.raytune import run_tune
from vizier_amd._src.raytune.run_tune import (
    ExperimenterConverter,
    SearchSpaceConverter,
)
