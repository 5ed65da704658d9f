"""Abstract experiment config.

Parity: /root/reference/maggy/config/lagom.py:22-34 (name / description /
hb_interval). ``hb_interval`` survives as the driver's metric-drain cadence
upper bound; with the shared-memory reporter there is no heartbeat socket.
"""
from abc import ABC


class LagomConfig(ABC):
    def __init__(self, name="maggyExperiment", description="", hb_interval=1):
        self.name = name
        self.description = description
        self.hb_interval = hb_interval
