"""Solver/analyzer parameters and defaults.

Parity with /root/reference/pkg/config/defaults.go.
"""

import math

# Tolerated percentile for SLOs
SLO_PERCENTILE = 0.95

# Multiplier of average of exponential distribution to attain percentile
SLO_MARGIN = -math.log(1 - SLO_PERCENTILE)

# Maximum number of requests in the queueing system as a multiple of max batch
MAX_QUEUE_TO_BATCH_RATIO = 10

# Accelerator transition penalty factor
ACCEL_PENALTY_FACTOR = 0.1

# Default service class
DEFAULT_SERVICE_CLASS_NAME = "Free"
DEFAULT_LOW_PRIORITY = 100
DEFAULT_HIGH_PRIORITY = 1
DEFAULT_SERVICE_CLASS_PRIORITY = DEFAULT_LOW_PRIORITY
