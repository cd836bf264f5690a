"""Public converters API (parity with vizier/pyvizier/converters)."""

from vizier_amd.converters.core import (
    PaddingSchedule,
    TrialToArrayConverter,
    pad_rows,
)
