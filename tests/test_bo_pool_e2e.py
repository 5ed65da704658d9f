"""GP and TPE through the REAL trial pool (async, multi-process)."""
from maggy_amd import Searchspace, experiment
from maggy_amd.config import HyperparameterOptConfig
from maggy_amd.optimizer.bayes import GP, TPE
from tests import _train_fns as fns


def _run(opt, exp_dir):
    sp = Searchspace(lr=("DOUBLE", [0.0, 0.12]))
    cfg = HyperparameterOptConfig(
        num_trials=14, optimizer=opt, searchspace=sp, direction="max",
        es_policy="none", num_workers=2, name="bo-pool")
    return experiment.lagom(fns.noisy_quadratic_fn, cfg)


def test_gp_pool_e2e(exp_dir):
    res = _run(GP(num_warmup_trials=6, random_fraction=0.15), exp_dir)
    assert res["num_trials"] == 14
    assert res["best_val"] > 0.8  # peak is 1.0 at lr=0.06


def test_tpe_pool_e2e(exp_dir):
    res = _run(TPE(num_warmup_trials=8, random_fraction=0.15), exp_dir)
    assert res["num_trials"] == 14
    assert res["best_val"] > 0.5
