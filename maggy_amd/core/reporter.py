"""Worker-side Reporter: the user-facing metric/log API inside train_fn.

Parity: /root/reference/maggy/core/reporter.py:30-170 — same ``broadcast``
validation (numeric metric, numeric monotone step) and the same contract
that a driver-initiated early stop surfaces as ``EarlyStopException`` raised
inside ``broadcast()``, and only after at least one metric was reported.

MI355X-native difference: metrics go into the lock-free shared-memory ring
(core/shm.py) instead of a heartbeat socket, and the stop signal is read
directly from the shared stop word tagged with the current trial id — no
socket round trip, no ``hb_interval`` latency.
"""
import threading
from datetime import datetime

from maggy_amd import constants, exceptions
from maggy_amd.core.shm import trial_tag


class Reporter:
    def __init__(self, ring=None, log_file=None, worker_id=0, print_fn=print):
        self.ring = ring
        self.worker_id = worker_id
        self.print_fn = print_fn
        self.lock = threading.RLock()
        self.metric = None
        self.step = -1
        self.stop = False  # local flag (driver-side use / tests)
        self.trial_id = None
        self._trial_tag = 0
        self.trial_log_file = None
        self.logs = ""
        self.log_file = log_file
        self.fd = open(log_file, "w") if log_file else None
        self.trial_fd = None

    # -- trial lifecycle ------------------------------------------------
    def set_trial_id(self, trial_id):
        with self.lock:
            self.trial_id = trial_id
            self._trial_tag = trial_tag(trial_id) if trial_id else 0

    def get_trial_id(self):
        with self.lock:
            return self.trial_id

    def init_logger(self, trial_log_file):
        self.trial_log_file = trial_log_file
        self.trial_fd = open(trial_log_file, "w")

    def close_logger(self):
        with self.lock:
            if self.trial_fd:
                self.trial_fd.close()
                self.trial_fd = None
            if self.fd:
                self.fd.close()
                self.fd = None

    # -- the user API ---------------------------------------------------
    def broadcast(self, metric, step=None):
        """Report a metric for the current trial.

        Raises :class:`EarlyStopException` if the driver flagged this trial
        for early stop (checked on every call; delivery latency is one
        driver event-loop iteration).
        """
        metric = self._to_scalar(metric)
        with self.lock:
            if step is None:
                step = self.step + 1
            if not isinstance(metric, constants.USER_FCT.NUMERIC_TYPES):
                raise exceptions.BroadcastMetricTypeError(metric)
            if not isinstance(step, constants.USER_FCT.NUMERIC_TYPES):
                raise exceptions.BroadcastStepTypeError(metric, step)
            if step < self.step:
                raise exceptions.BroadcastStepValueError(metric, step, self.step)
            self.step = step
            self.metric = metric
            if self.ring is not None:
                self.ring.push(self._trial_tag, step, float(metric))
            # early-stop check: shared stop word tagged with our trial
            stopped = self.stop
            if (
                not stopped
                and self.ring is not None
                and self._trial_tag != 0
                and self.ring.read_stop_word() == self._trial_tag
            ):
                stopped = True
            if stopped:
                raise exceptions.EarlyStopException(metric)

    @staticmethod
    def _to_scalar(metric):
        """Allow broadcast() to take a torch tensor: a 0-d tensor reads its
        item; a larger CUDA tensor is mean-reduced by the HIP reduction
        kernel (reference call-site N8)."""
        if type(metric).__module__.startswith("torch"):
            if metric.numel() == 1:
                return float(metric.detach().item())
            from maggy_amd.ops import metric_mean

            return metric_mean(metric.detach())
        return metric

    def log(self, log_msg, jupyter=False):
        """Log to the worker logfile (and the trial logfile when a trial is
        active).  When a ``log_sink`` is attached (the worker's control
        pipe), accumulated jupyter-stream text is flushed through it at
        most once per second — live log streaming to the driver mid-trial
        (the reference shipped logs with every heartbeat,
        rpc.py:723-726)."""
        flush = None
        with self.lock:
            msg = "{} ({}): {}\n".format(
                datetime.now().isoformat(), self.worker_id, log_msg
            )
            try:
                if self.trial_fd:
                    self.trial_fd.write(msg)
                if jupyter:
                    self.logs += "{}: {}\n".format(self.worker_id, log_msg)
                else:
                    if self.fd:
                        self.fd.write(msg)
                    self.print_fn(msg.rstrip("\n"))
            except (IOError, ValueError, AttributeError):
                pass
            sink = getattr(self, "log_sink", None)
            if sink is not None and self.logs:
                import time as _time

                now = _time.time()
                if now - getattr(self, "_last_sink", 0.0) >= 1.0:
                    self._last_sink = now
                    flush, self.logs = self.logs, ""
        if flush is not None:
            try:
                sink(flush)
            except Exception:
                pass

    # -- driver/test helpers --------------------------------------------
    def get_data(self):
        with self.lock:
            logs = self.logs
            self.logs = ""
            return self.metric, self.step, logs

    def reset(self):
        with self.lock:
            self.metric = None
            self.step = -1
            self.stop = False
            self.trial_id = None
            self._trial_tag = 0
            if self.fd:
                self.fd.flush()
            if self.trial_fd:
                self.trial_fd.close()
                self.trial_fd = None
            self.trial_log_file = None

    def early_stop(self):
        """Set the local stop flag (takes effect only after >=1 metric was
        reported, parity reference reporter.py:159-162)."""
        with self.lock:
            if self.metric is not None:
                self.stop = True
