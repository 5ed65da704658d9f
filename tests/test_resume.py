"""Experiment resume from persisted trial.json records."""
import os

from maggy_amd import Searchspace, experiment
from maggy_amd.config import HyperparameterOptConfig
from maggy_amd.core.driver import OptimizationDriver
from tests import _train_fns as fns


def _run_dir(exp_dir):
    app = [d for d in os.listdir(exp_dir) if d.startswith("application_")][0]
    runs = sorted(os.listdir(os.path.join(exp_dir, app)), key=int)
    return os.path.join(exp_dir, app, runs[0]), app


def test_resume_randomsearch(exp_dir):
    sp = Searchspace(lr=("DOUBLE", [0.01, 0.1]))
    cfg = HyperparameterOptConfig(
        num_trials=4, optimizer="randomsearch", searchspace=sp,
        direction="max", es_policy="none", num_workers=2, name="part1")
    res1 = experiment.lagom(fns.metric_eq_lr, cfg)
    assert res1["num_trials"] == 4
    run1, app = _run_dir(exp_dir)

    # resume: driver preloads the 4 finalized trials, controller skips
    # them, and only 2 more run
    cfg2 = HyperparameterOptConfig(
        num_trials=6, optimizer="randomsearch", searchspace=sp,
        direction="max", es_policy="none", num_workers=2, name="part2")
    d = OptimizationDriver(cfg2, app_id=app)
    assert d.resume_from(run1) == 4
    res2 = d.run_experiment(fns.metric_eq_lr)
    assert res2["num_trials"] == 6
    # best over ALL six trials (old ones included in the result)
    assert res2["best_val"] >= res1["best_val"]


def test_resume_unsupported_controller(exp_dir):
    sp = Searchspace(lr=("DOUBLE", [0.01, 0.1]))
    cfg = HyperparameterOptConfig(
        num_trials=4, optimizer="randomsearch", searchspace=sp,
        es_policy="none", num_workers=1, name="p1")
    experiment.lagom(fns.metric_eq_lr, cfg)
    run1, app = _run_dir(exp_dir)

    sp2 = Searchspace(lr=("DISCRETE", [0.01, 0.1]))
    cfg2 = HyperparameterOptConfig(
        num_trials=2, optimizer="gridsearch", searchspace=sp2,
        es_policy="none", num_workers=1, name="p2")
    d = OptimizationDriver(cfg2, app_id=app)
    d.resume_from(run1)
    import pytest

    # GridSearch has no on_resume -> refuse rather than re-run configs
    with pytest.raises(NotImplementedError):
        d.run_experiment(fns.metric_eq_lr)
