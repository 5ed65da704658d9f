"""Hyperband pruner through the REAL trial pool: exercises the IDLE
requeue path (workers wait while a rung completes) end to end."""
import json
import os

from maggy_amd import Searchspace, experiment
from maggy_amd.config import HyperparameterOptConfig
from maggy_amd.optimizer import RandomSearch
from tests import _train_fns as fns


def test_hyperband_pool_e2e(exp_dir):
    sp = Searchspace(lr=("DOUBLE", [0.01, 0.1]))
    opt = RandomSearch(pruner="hyperband",
                       pruner_kwargs=dict(min_budget=1, max_budget=4,
                                          eta=2, n_iterations=1))
    cfg = HyperparameterOptConfig(
        num_trials=7, optimizer=opt, searchspace=sp, direction="max",
        es_policy="none", num_workers=3, name="hb-pool")
    res = experiment.lagom(fns.budgeted_fn, cfg)
    # bracket: [4, 2, 1] configs at budgets [1, 2, 4] -> 7 trials
    assert res["num_trials"] == 7
    assert opt.pruner.finished()

    app = [d for d in os.listdir(exp_dir) if d.startswith("application_")][0]
    run_dir = os.path.join(exp_dir, app,
                           sorted(os.listdir(os.path.join(exp_dir, app)))[0])
    budgets = []
    for td in os.listdir(run_dir):
        hp = os.path.join(run_dir, td, ".hparams.json")
        if os.path.isdir(os.path.join(run_dir, td)) and os.path.exists(hp):
            budgets.append(json.load(open(hp)).get("budget"))
    assert sorted(budgets) == [1, 1, 1, 1, 2, 2, 4]
    # pruner.log artifact exists
    assert os.path.exists(os.path.join(run_dir, "pruner.log"))


def test_hyperband_pool_survives_trial_error(exp_dir):
    """A train_fn that raises must not stall its bracket: the slot is freed
    and re-run, and the experiment completes."""
    sp = Searchspace(lr=("DOUBLE", [0.01, 0.1]))
    opt = RandomSearch(pruner="hyperband",
                       pruner_kwargs=dict(min_budget=1, max_budget=4,
                                          eta=2, n_iterations=1))
    cfg = HyperparameterOptConfig(
        num_trials=7, optimizer=opt, searchspace=sp, direction="max",
        es_policy="none", num_workers=2, name="hb-err")
    res = experiment.lagom(fns.fails_once_fn, cfg)
    # 7 successful trials despite the first one erroring
    assert res["num_trials"] == 7
    assert opt.pruner.finished()
