"""Optimizer registry.

Parity: /root/reference/maggy/core/experiment_driver/optimization_driver.py
:49-57 — registry strings {randomsearch, asha, tpe, gp, none, gridsearch}
or a user-provided AbstractOptimizer instance.
"""
from maggy_amd.optimizer.abstract import AbstractOptimizer  # noqa: F401
from maggy_amd.optimizer.asha import Asha  # noqa: F401
from maggy_amd.optimizer.gridsearch import GridSearch  # noqa: F401
from maggy_amd.optimizer.randomsearch import RandomSearch  # noqa: F401
from maggy_amd.optimizer.singlerun import SingleRun  # noqa: F401

__all__ = [
    "AbstractOptimizer", "Asha", "GridSearch", "RandomSearch", "SingleRun",
    "resolve_controller",
]


def _gp():
    from maggy_amd.optimizer.bayes.gp import GP  # lazy: pulls in sklearn

    return GP()


def _tpe():
    from maggy_amd.optimizer.bayes.tpe import TPE  # lazy: pulls in scipy

    return TPE()


def _registry():
    return {
        "randomsearch": RandomSearch,
        "asha": Asha,
        "gridsearch": GridSearch,
        "none": SingleRun,
        "gp": _gp,
        "tpe": _tpe,
    }


def resolve_controller(optimizer, driver):
    """Instantiate/wire the controller from a registry string or instance
    and attach searchspace / num_trials / stores / direction."""
    if optimizer is None:
        controller = SingleRun()
    elif isinstance(optimizer, str):
        key = optimizer.lower()
        reg = _registry()
        if key not in reg:
            raise ValueError(
                "Unknown optimizer '{}'; choose one of {} or pass an "
                "AbstractOptimizer instance".format(
                    optimizer, sorted(reg.keys())))
        controller = reg[key]()
    elif isinstance(optimizer, AbstractOptimizer):
        controller = optimizer
    else:
        raise ValueError(
            "optimizer must be a string, None or an AbstractOptimizer, got "
            "{}".format(type(optimizer)))
    controller.searchspace = driver.searchspace
    controller.num_trials = driver.num_trials
    controller.trial_store = driver._trial_store
    controller.final_store = driver._final_store
    controller.direction = driver.direction
    return controller
