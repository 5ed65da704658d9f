"""Abstract experiment config.

Parity: /root/reference/maggy/config/lagom.py:22-34 (name / description /
hb_interval). ``hb_interval`` is VESTIGIAL: it is accepted for source
compatibility with reference configs but has no effect — the reference used
it as the executor heartbeat-socket period (rpc.py:716-737); here metrics
stream through shared-memory rings drained every event-loop tick, so there
is no heartbeat cadence to configure.
"""
from abc import ABC


class LagomConfig(ABC):
    def __init__(self, name="maggyExperiment", description="", hb_interval=1):
        self.name = name
        self.description = description
        self.hb_interval = hb_interval
