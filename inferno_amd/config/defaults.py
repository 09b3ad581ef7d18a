"""Global tuning constants. Ref: pkg/config/defaults.go:12-37, pkg/config/config.go:4-41."""
from __future__ import annotations

import math
from enum import Enum

# Tolerated percentile for SLOs (ref defaults.go:13)
SLO_PERCENTILE = 0.95

# Multiplier of average of exponential distribution to attain percentile (defaults.go:16)
SLO_MARGIN = -math.log(1.0 - SLO_PERCENTILE)

# Maximum number of requests in queueing system as multiple of max batch size (defaults.go:19)
MAX_QUEUE_TO_BATCH_RATIO = 10

# Accelerator transition penalty factor (defaults.go:22)
ACCEL_PENALTY_FACTOR = 0.1

# LDS-resident chain-state threshold (max batch size N per cell that fits
# the HIP kernel's LDS geometry, WVA_MAX_N). N is NOT capped — matching the
# reference's uncapped sizing (allocation.go:80-86) — cells above this spill
# their chain geometry to a global-memory slab on the GPU backend.
MAX_BATCH_STATES = 8192

# Service-class defaults (defaults.go:24-33)
DEFAULT_SERVICE_CLASS_NAME = "Free"
DEFAULT_LOW_PRIORITY = 100
DEFAULT_HIGH_PRIORITY = 1
DEFAULT_SERVICE_CLASS_PRIORITY = DEFAULT_LOW_PRIORITY


class SaturationPolicy(str, Enum):
    """Allocation policy under saturated capacity. Ref: pkg/config/config.go:6-11."""

    NONE = "None"
    PRIORITY_EXHAUSTIVE = "PriorityExhaustive"
    PRIORITY_ROUND_ROBIN = "PriorityRoundRobin"
    ROUND_ROBIN = "RoundRobin"

    @classmethod
    def parse(cls, s: str) -> "SaturationPolicy":
        try:
            return cls(s)
        except ValueError:
            return cls.NONE


DEFAULT_SATURATION_POLICY = SaturationPolicy.NONE
