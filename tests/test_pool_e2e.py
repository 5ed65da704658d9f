"""End-to-end trial-pool experiments on CPU (multi-process, no GPU)."""
import json
import os

import pytest

from maggy_amd import Searchspace, experiment
from maggy_amd.config import BaseConfig, HyperparameterOptConfig
from tests import _train_fns as fns


def _artifact_dir(exp_dir):
    app = [d for d in os.listdir(exp_dir) if d.startswith("application_")][0]
    run = sorted(os.listdir(os.path.join(exp_dir, app)))[0]
    return os.path.join(exp_dir, app, run)


def test_randomsearch_e2e(exp_dir):
    sp = Searchspace(lr=("DOUBLE", [0.01, 0.1]))
    cfg = HyperparameterOptConfig(
        num_trials=5, optimizer="randomsearch", searchspace=sp,
        direction="max", es_policy="none", num_workers=2, name="rs-e2e")
    res = experiment.lagom(fns.quick_fn, cfg)
    assert res["num_trials"] == 5
    assert res["best_val"] >= res["worst_val"]
    d = _artifact_dir(exp_dir)
    # experiment-level artifacts
    assert os.path.exists(os.path.join(d, "result.json"))
    assert os.path.exists(os.path.join(d, "maggy.json"))
    assert os.path.exists(os.path.join(d, "maggy.log"))
    assert os.path.exists(os.path.join(d, "optimizer.log"))
    meta = json.load(open(os.path.join(d, "maggy.json")))
    assert meta["status"] == "FINISHED"
    assert meta["experiment_type"] == "RandomSearch"
    # per-trial artifact tree
    trial_dirs = [x for x in os.listdir(d)
                  if os.path.isdir(os.path.join(d, x))]
    assert len(trial_dirs) == 5
    for td in trial_dirs:
        for f in (".hparams.json", ".outputs.json", ".metric", "trial.json",
                  "output.log"):
            assert os.path.exists(os.path.join(d, td, f)), (td, f)
        tj = json.load(open(os.path.join(d, td, "trial.json")))
        assert tj["status"] == "FINALIZED"
        assert tj["final_metric"] is not None
        assert len(tj["metric_history"]) == 4


def test_asha_e2e(exp_dir):
    sp = Searchspace(lr=("DOUBLE", [0.01, 0.1]))
    cfg = HyperparameterOptConfig(
        num_trials=16, optimizer="asha", searchspace=sp,
        direction="max", es_policy="none", num_workers=3, name="asha-e2e")
    res = experiment.lagom(fns.budgeted_fn, cfg)
    assert res["num_trials"] >= 4
    d = _artifact_dir(exp_dir)
    budgets = set()
    for td in os.listdir(d):
        hp = os.path.join(d, td, ".hparams.json")
        if os.path.isdir(os.path.join(d, td)) and os.path.exists(hp):
            budgets.add(json.load(open(hp)).get("budget"))
    assert 4 in budgets  # a trial reached the max rung


def test_dict_return_and_outputs(exp_dir):
    sp = Searchspace(lr=("DOUBLE", [0.01, 0.1]))
    cfg = HyperparameterOptConfig(
        num_trials=2, optimizer="randomsearch", searchspace=sp,
        es_policy="none", num_workers=1, name="dict-e2e")
    res = experiment.lagom(fns.returns_dict_fn, cfg)
    assert res["best_val"] == 0.5
    d = _artifact_dir(exp_dir)
    td = [x for x in os.listdir(d) if os.path.isdir(os.path.join(d, x))][0]
    out = json.load(open(os.path.join(d, td, ".outputs.json")))
    assert out["Metric"] == 0.5 and out["aux"] == 1.0


def test_trainfn_exception_marks_error_and_continues(exp_dir):
    sp = Searchspace(boom=("DISCRETE", [0, 1]))
    cfg = HyperparameterOptConfig(
        num_trials=4, optimizer="gridsearch", searchspace=sp,
        es_policy="none", num_workers=2, name="err-e2e")
    res = experiment.lagom(fns.crashing_fn, cfg)
    # the grid is {0,1}: one good, one crashing; experiment completes
    assert res.get("num_trials", 0) >= 1
    assert res["best_val"] == 1.0


def test_worker_death_respawn(exp_dir):
    sp = Searchspace(die=("DISCRETE", [0, 1]))
    cfg = HyperparameterOptConfig(
        num_trials=2, optimizer="gridsearch", searchspace=sp,
        es_policy="none", num_workers=1, name="death-e2e")
    res = experiment.lagom(fns.suicide_fn, cfg)
    # the dying trial is re-assigned to the respawned worker; the fresh
    # process no longer has MAGGY_TEST_DIED set, so it completes
    assert res["num_trials"] == 2
    assert res["best_val"] == 2.0


def test_base_config_single_run(exp_dir):
    cfg = BaseConfig(name="single", hparams={"x": 1})
    res = experiment.lagom(fns.single_run_fn, cfg)
    assert res["Metric"] == 3.0 and res["extra"] == 7
    d = _artifact_dir(exp_dir)
    assert os.path.exists(os.path.join(d, "single_run", ".outputs.json"))


def test_running_guard():
    from maggy_amd.experiment import experiment as exp_mod

    exp_mod.running = True
    try:
        with pytest.raises(RuntimeError):
            experiment.lagom(fns.quick_fn, BaseConfig())
    finally:
        exp_mod.running = False
