"""Typed exceptions.

Parity: /root/reference/maggy/core/exceptions.py:22-121. Same user-visible
exception names and trigger conditions; messages are our own.
"""


class EarlyStopException(Exception):
    """Raised inside ``reporter.broadcast()`` when the driver decided to stop
    the running trial (median rule / controller decision).

    The trial executor catches this, finalizes the trial with the last metric
    and pulls the next trial.
    """

    def __init__(self, metric=None):
        super().__init__("Trial early-stopped by the experiment driver.")
        self.metric = metric


class NotSupportedError(Exception):
    """An argument combination is not supported by this framework."""

    def __init__(self, category, value, suggestion=""):
        msg = "{} '{}' is not supported. {}".format(category, value, suggestion)
        super().__init__(msg)


class BadArgumentsError(Exception):
    """An API was called with inconsistent arguments."""

    def __init__(self, func, hint=""):
        super().__init__("Bad arguments for {}. {}".format(func, hint))


class ReturnTypeError(Exception):
    """The training function returned a value of unusable type."""

    def __init__(self, optimization_key, return_val):
        super().__init__(
            "Training function returned {} (type {}); expected a number or a "
            "dict containing the optimization key '{}'.".format(
                return_val, type(return_val).__name__, optimization_key
            )
        )


class MetricTypeError(Exception):
    """The optimization metric inside the returned dict is not numeric."""

    def __init__(self, optimization_key, value):
        super().__init__(
            "Optimization metric '{}' has non-numeric value {} (type {}).".format(
                optimization_key, value, type(value).__name__
            )
        )


class BroadcastMetricTypeError(Exception):
    """``reporter.broadcast`` called with a non-numeric metric."""

    def __init__(self, metric):
        super().__init__(
            "broadcast() metric must be numeric, got {} (type {}).".format(
                metric, type(metric).__name__
            )
        )


class BroadcastStepTypeError(Exception):
    """``reporter.broadcast`` called with a non-numeric step."""

    def __init__(self, metric, step):
        super().__init__(
            "broadcast() step must be numeric, got {} (type {}).".format(
                step, type(step).__name__
            )
        )


class BroadcastStepValueError(Exception):
    """``reporter.broadcast`` called with a non-monotone step."""

    def __init__(self, metric, step, prev_step):
        super().__init__(
            "broadcast() steps must be monotonically increasing: got step {} "
            "after step {}.".format(step, prev_step)
        )


class WorkerCrashError(Exception):
    """A trial-pool worker process died (nonzero exit / signal)."""

    def __init__(self, worker_id, detail):
        super().__init__(
            "Trial worker(s) {} died: {}".format(worker_id, detail)
        )
        self.worker_id = worker_id
        self.detail = detail


class HipExtensionMissingError(ImportError):
    """The compiled HIP extension is required (GPU present) but not importable."""

    def __init__(self, detail=""):
        super().__init__(
            "maggy_amd HIP extension (_maggy_hip) is not built/loadable but a "
            "GPU is present - refusing silent eager fallback. Build it with "
            "`python -m maggy_amd.ops.build`. {}".format(detail)
        )
