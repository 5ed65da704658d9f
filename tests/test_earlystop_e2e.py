"""End-to-end median early stop through the trial pool: underperforming
trials must receive EarlyStopException mid-trial via the shared stop word."""
import json
import os

from maggy_amd import Searchspace, experiment
from maggy_amd.config import HyperparameterOptConfig
from tests import _train_fns as fns


def test_median_early_stop_e2e(exp_dir):
    # grid pops in order: the two strong trials (10, 20) finalize first,
    # then the weak ones (1, 2) fall below the median and get stopped
    sp = Searchspace(level=("DISCRETE", [10, 20, 1, 2]))
    cfg = HyperparameterOptConfig(
        num_trials=4, optimizer="gridsearch", searchspace=sp,
        direction="max", es_policy="median", es_interval=1, es_min=1,
        num_workers=2, name="es-e2e")
    res = experiment.lagom(fns.slow_fn_for_earlystop, cfg)
    assert res["num_trials"] == 4
    assert res["early_stopped"] >= 1
    assert res["best_val"] == 20

    app = [d for d in os.listdir(exp_dir) if d.startswith("application_")][0]
    run_dir = os.path.join(exp_dir, app,
                           sorted(os.listdir(os.path.join(exp_dir, app)))[0])
    short_histories = 0
    for td in os.listdir(run_dir):
        tj = os.path.join(run_dir, td, "trial.json")
        if os.path.isdir(os.path.join(run_dir, td)) and os.path.exists(tj):
            t = json.load(open(tj))
            if t["early_stop"]:
                assert len(t["metric_history"]) < 60
                short_histories += 1
    assert short_histories >= 1
