"""No-op early-stop policy (parity: /root/reference/maggy/earlystop/nostop.py)."""
from maggy_amd.earlystop.abstract import AbstractEarlyStop


class NoStoppingRule(AbstractEarlyStop):
    @staticmethod
    def earlystop_check(to_check, finalized_trials, direction):
        return None
