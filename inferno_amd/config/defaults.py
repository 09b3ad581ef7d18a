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

# Hard cap on queueing-chain batch states (max batch size N per cell).
# The reference has no cap (its chain is an O(11*N) sequential recurrence);
# this build caps N at the HIP kernel's LDS budget (WVA_MAX_N) and applies
# the same cap on every backend so CPU and GPU sweeps agree exactly.
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
