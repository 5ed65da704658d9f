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


def lagom(train_fn, config=None):
    """Launch a maggy experiment; returns the result dict."""
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
        result = _lagom_driver(config, train_fn)
        return result
    finally:
        running = False


@functools.singledispatch
def _lagom_driver(config, train_fn):
    raise TypeError("Unsupported config type: {}".format(type(config)))


@_lagom_driver.register(BaseConfig)
def _(config, train_fn):
    from maggy_amd.core.base_driver import BaseDriver

    return BaseDriver(config).run_experiment(train_fn)


@_lagom_driver.register(HyperparameterOptConfig)
def _(config, train_fn):
    from maggy_amd.core.driver import OptimizationDriver

    return OptimizationDriver(config).run_experiment(train_fn)


@_lagom_driver.register(AblationConfig)
def _(config, train_fn):
    from maggy_amd.ablation.driver import AblationDriver

    return AblationDriver(config).run_experiment(train_fn)


@_lagom_driver.register(TorchDistributedConfig)
def _(config, train_fn):
    from maggy_amd.parallel.driver import TorchDistributedTrainingDriver

    return TorchDistributedTrainingDriver(config).run_experiment(train_fn)
