"""Early-stop policy interface.

Parity: /root/reference/maggy/earlystop/abstractearlystop.py:23-42.
"""
from abc import ABC, abstractmethod


class AbstractEarlyStop(ABC):
    @staticmethod
    @abstractmethod
    def earlystop_check(to_check, finalized_trials, direction):
        """Return the trial_id to stop, or None."""
