"""Promoted trials continue from the parent checkpoint (ASHA)."""
from maggy_amd import Searchspace, experiment
from maggy_amd.config import HyperparameterOptConfig
from tests import _train_fns as fns


def test_asha_checkpoint_continuation(exp_dir):
    sp = Searchspace(lr=("DOUBLE", [0.01, 0.1]))
    cfg = HyperparameterOptConfig(
        num_trials=16, optimizer="asha", searchspace=sp, direction="max",
        es_policy="none", num_workers=2, name="cont")
    res = experiment.lagom(fns.continuation_fn, cfg)
    # rung ladder budgets 1 -> 2 -> 4; with continuation the final-rung
    # trial has 1 (parent) + 2 (parent) + 4 = 7 cumulative steps
    assert res["best_val"] == 7.0
