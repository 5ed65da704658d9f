"""ASHA resume: rung reconstruction from persisted trial records."""
import os

from maggy_amd import Searchspace, experiment
from maggy_amd.config import HyperparameterOptConfig
from maggy_amd.core.driver import OptimizationDriver
from maggy_amd.optimizer import Asha
from tests import _train_fns as fns


def test_asha_resume_rungs(exp_dir):
    sp = Searchspace(lr=("DOUBLE", [0.01, 0.1]))
    cfg = HyperparameterOptConfig(
        num_trials=16, optimizer="asha", searchspace=sp, direction="max",
        es_policy="none", num_workers=2, name="a1")
    res1 = experiment.lagom(fns.budgeted_fn, cfg)
    n1 = res1["num_trials"]

    app = [d for d in os.listdir(exp_dir) if d.startswith("application_")][0]
    run1 = os.path.join(exp_dir, app,
                        sorted(os.listdir(os.path.join(exp_dir, app)))[0])

    # resume into a fresh ASHA: rungs rebuilt; since the previous
    # experiment completed its bracket (max rung populated), the resumed
    # controller must terminate immediately with zero new trials
    opt = Asha(reduction_factor=2, resource_min=1, resource_max=4)
    cfg2 = HyperparameterOptConfig(
        num_trials=16, optimizer=opt, searchspace=sp, direction="max",
        es_policy="none", num_workers=2, name="a2")
    d = OptimizationDriver(cfg2, app_id=app)
    assert d.resume_from(run1) == n1
    res2 = d.run_experiment(fns.budgeted_fn)
    assert 2 in opt.rungs and len(opt.rungs[2]) >= 1
    # promoted bookkeeping was reconstructed
    assert opt.promoted.get(0) and opt.promoted.get(1)
    # no new trials were run: result covers exactly the resumed set
    assert res2["num_trials"] == n1
