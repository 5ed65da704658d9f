"""``experiment.lagom(train_fn, config)`` — the single user entry point.

Parity: /root/reference/maggy/experiment/experiment.py:21 and the
singledispatch driver selection of experiment_pyspark.py:82-146.  "Lagom"
keeps its meaning: the same oblivious training function runs as a single
run (BaseConfig), an async hyperparameter search (HyperparameterOptConfig),
an ablation study (AblationConfig) or distributed data-parallel training
(TorchDistributedConfig) — here on a single-node MI355X trial pool instead
of a Spark cluster.
"""
import functools

from maggy_amd.config import (
    AblationConfig,
    BaseConfig,
    HyperparameterOptConfig,
    LagomConfig,
    TorchDistributedConfig,
)

_APP_ID = None
_RUN_ID = 1
running = False


def lagom(train_fn, config=None, progress=None):
    """Launch a maggy experiment; returns the result dict.

    ``progress``: optional callable(status_string, new_log_text) invoked
    from the driver's event loop every few seconds DURING the run — the
    reference streamed the same snapshot to Jupyter over LOG requests
    (/root/reference/maggy/core/rpc.py:490-502); here it is an in-process
    callback, no socket.
    """
    global running, _APP_ID, _RUN_ID
    if running:
        raise RuntimeError("An experiment is currently running.")
    if config is None:
        config = BaseConfig(name="no-config-default")
    if not isinstance(config, LagomConfig):
        raise TypeError(
            "config must be a LagomConfig subclass, got {}".format(
                type(config)))
    try:
        running = True
        result = _lagom_driver(config, train_fn, progress)
        return result
    finally:
        running = False


class LagomHandle:
    """Handle for a lagom experiment running on a background thread:
    poll ``get_logs()``/``done()`` mid-run, ``result()`` to join."""

    def __init__(self, thread, driver_box):
        self._thread = thread
        self._driver_box = driver_box
        self._result = None
        self._exc = None

    def done(self):
        return not self._thread.is_alive()

    def get_logs(self):
        """(status_string, new_log_text) snapshot; safe mid-run."""
        driver = self._driver_box.get("driver")
        if driver is None or not hasattr(driver, "get_logs"):
            return "", ""
        return driver.get_logs()

    def result(self, timeout=None):
        self._thread.join(timeout)
        if self._thread.is_alive():
            raise TimeoutError("experiment still running")
        if self._exc is not None:
            raise self._exc
        return self._result


def lagom_async(train_fn, config=None):
    """Start ``lagom`` on a background thread and return a LagomHandle —
    the Jupyter usage pattern: kick off the experiment, keep polling
    ``handle.get_logs()`` for live progress, then ``handle.result()``."""
    import threading

    box = {}
    handle_holder = {}

    def _run():
        h = handle_holder["h"]
        try:
            # no progress callback: _emit_progress would DRAIN the log
            # stream into it, racing the handle's own get_logs(); the
            # driver is published to the box by _lagom_driver regardless
            h._result = lagom(train_fn, config)
        except BaseException as e:  # surfaced via result()
            h._exc = e

    t = threading.Thread(target=_run, daemon=True)
    h = LagomHandle(t, box)
    handle_holder["h"] = h
    _DRIVER_BOX.append(box)
    t.start()
    return h


# driver registry so lagom_async can reach the live driver's get_logs()
_DRIVER_BOX = []


def _publish_driver(driver):
    while _DRIVER_BOX:
        _DRIVER_BOX.pop()["driver"] = driver


@functools.singledispatch
def _lagom_driver(config, train_fn, progress=None):
    raise TypeError("Unsupported config type: {}".format(type(config)))


@_lagom_driver.register(BaseConfig)
def _(config, train_fn, progress=None):
    from maggy_amd.core.base_driver import BaseDriver

    driver = BaseDriver(config)
    _publish_driver(driver)
    return driver.run_experiment(train_fn)


@_lagom_driver.register(HyperparameterOptConfig)
def _(config, train_fn, progress=None):
    from maggy_amd.core.driver import OptimizationDriver

    driver = OptimizationDriver(config)
    driver.progress_cb = progress
    _publish_driver(driver)
    return driver.run_experiment(train_fn)


@_lagom_driver.register(AblationConfig)
def _(config, train_fn, progress=None):
    from maggy_amd.ablation.driver import AblationDriver

    driver = AblationDriver(config)
    driver.progress_cb = progress
    _publish_driver(driver)
    return driver.run_experiment(train_fn)


@_lagom_driver.register(TorchDistributedConfig)
def _(config, train_fn, progress=None):
    from maggy_amd.parallel.driver import TorchDistributedTrainingDriver

    driver = TorchDistributedTrainingDriver(config)
    _publish_driver(driver)
    return driver.run_experiment(train_fn)
