"""Pruner interface.

Parity: /root/reference/maggy/pruner/abstractpruner.py:38-95 — the pruner
decides which config runs next and at what budget; the optimizer routes
through ``pruning_routine()`` and reports the mapping original->new trial
ids via ``report_trial()``.
"""
import time
from abc import ABC, abstractmethod


class AbstractPruner(ABC):
    def __init__(self, trial_metric_getter):
        """:param trial_metric_getter: callable(trial_ids) -> {id: metric}
        in MIN convention (driver metrics negated for max experiments)."""
        self.trial_metric_getter = trial_metric_getter
        self._log_fd = None

    def _initialize(self, exp_dir=None):
        if exp_dir is not None:
            self._log_fd = open(exp_dir + "/pruner.log", "w")
        self.initialize()

    def _close_log(self):
        if self._log_fd is not None:
            self._log_fd.close()
            self._log_fd = None

    def _log(self, msg):
        if self._log_fd is not None and not self._log_fd.closed:
            self._log_fd.write(
                "{}: {}\n".format(time.strftime("%H:%M:%S"), msg))
            self._log_fd.flush()

    def initialize(self):
        pass

    @abstractmethod
    def pruning_routine(self):
        """Return {"trial_id": parent_or_None, "budget": b} | "IDLE" | None."""

    @abstractmethod
    def report_trial(self, original_trial_id, new_trial_id):
        ...

    @abstractmethod
    def finished(self):
        ...

    @abstractmethod
    def num_trials(self):
        ...

    def name(self):
        return str(self.__class__.__name__)
