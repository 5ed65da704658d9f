"""Framework-wide constants.

Parity: /root/reference/maggy/constants.py:23-27 (allowed return and metric
types for the user training function).
"""
import numpy as np


class USER_FCT:
    """Types a user training function may return / report."""

    # a training function may return a scalar metric or a dict of metrics
    RETURN_TYPES = (float, int, dict, np.number)
    # a reported metric must be numeric
    NUMERIC_TYPES = (float, int, np.number)


class SCHEDULER:
    """Trial-pool scheduler knobs (this framework's control plane)."""

    # metric ring: records per worker ring buffer (power of two)
    RING_SLOTS = 4096
    # seconds the driver event loop blocks in connection.wait
    POLL_TIMEOUT = 0.05
    # seconds to wait for worker processes to join on shutdown
    JOIN_TIMEOUT = 30.0
