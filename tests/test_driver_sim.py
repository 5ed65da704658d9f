"""Deterministic scheduler state-machine tests: the OptimizationDriver's
message handlers driven by a fake pool (no processes, no timing) — the
fake-executor harness SURVEY.md §4 calls for.

Covers: REG first assignment, FINAL -> next-suggestion hand-off, metric
digestion + median early stop -> stop-word flagging, worker-death
reassignment (BLACK), IDLE requeue, experiment-done transitions.
"""
import pytest

from maggy_amd import Searchspace
from maggy_amd.config import HyperparameterOptConfig
from maggy_amd.core import messages as M
from maggy_amd.core.driver import OptimizationDriver
from maggy_amd.core.shm import trial_tag
from maggy_amd.trial import Trial


class FakeWorker:
    def __init__(self, worker_id):
        self.worker_id = worker_id
        self.trial_id = None
        self.registered = False
        self.respawns = 0
        self.process = None
        self.ring = self
        self.conn = None
        # ring-api surface
        self.stop_word = 0
        self.assigned = []

    def clear_stop(self):
        self.stop_word = 0

    def set_stop(self, tag):
        self.stop_word = tag

    def drain(self):
        return []


class FakePool:
    def __init__(self, n):
        self.workers = [FakeWorker(i) for i in range(n)]

    def assign(self, w, trial):
        w.trial_id = trial.trial_id
        w.assigned.append((trial.trial_id, dict(trial.params)))

    def request_stop(self, trial_id):
        for w in self.workers:
            if w.trial_id == trial_id:
                w.set_stop(trial_tag(trial_id))
                return True
        return False

    def reap(self):
        return [w for w in self.workers
                if w.process is not None and not w.process.is_alive()]


@pytest.fixture
def driver(exp_dir):
    sp = Searchspace(lr=("DOUBLE", [0.0, 1.0]))
    cfg = HyperparameterOptConfig(
        num_trials=4, optimizer="randomsearch", searchspace=sp,
        direction="max", es_policy="median", es_interval=1, es_min=0,
        num_workers=2, name="sim")
    d = OptimizationDriver(cfg)
    d.pool = FakePool(2)
    d.controller._initialize(exp_dir=d.log_dir)
    return d


def reg(d, wid):
    w = d.pool.workers[wid]
    w.registered = True
    if not d._assign_pending(w):
        d._assign_next(w)
    return w


def finalize(d, w, metric, early=False):
    tid = w.trial_id
    d._handle_final(w, (M.FINAL, w.worker_id, tid, metric, 0.5, early, ""))
    return tid


def test_reg_assigns_and_final_hands_off(driver):
    w0 = reg(driver, 0)
    w1 = reg(driver, 1)
    assert w0.trial_id is not None and w1.trial_id is not None
    assert w0.trial_id != w1.trial_id
    t0 = finalize(driver, w0, 1.0)
    # FINAL immediately handed the worker the 3rd trial
    assert w0.trial_id is not None and w0.trial_id != t0
    finalize(driver, w1, 2.0)
    finalize(driver, w0, 3.0)
    finalize(driver, w1, 0.5)
    # 4 trials done -> controller exhausted -> experiment done
    assert driver.experiment_done
    assert driver.result["num_trials"] == 4
    assert driver.result["best_val"] == 3.0
    assert driver.result["worst_val"] == 0.5


def test_metric_digestion_and_early_stop(driver):
    w0 = reg(driver, 0)
    w1 = reg(driver, 1)
    # finalize one strong trial so the median rule has history
    strong = driver.get_trial(w0.trial_id)
    for s in range(3):
        driver._handle_metrics([(w0, trial_tag(strong.trial_id), s, 10.0)])
    finalize(driver, w0, 10.0)
    # now stream weak metrics for w1's trial -> early-stop flag + stop word
    weak = driver.get_trial(w1.trial_id)
    driver._handle_metrics([(w1, trial_tag(weak.trial_id), 1, 0.1)])
    assert weak.get_early_stop()
    assert w1.stop_word == trial_tag(weak.trial_id)
    # metric history was recorded and deduped
    driver._handle_metrics([(w1, trial_tag(weak.trial_id), 1, 0.2)])
    assert weak.metric_history == [0.1]


def test_worker_death_reassigns_trial(driver, monkeypatch):
    w0 = reg(driver, 0)
    lost = w0.trial_id

    class DeadProc:
        exitcode = 9

        @staticmethod
        def is_alive():
            return False

    w0.process = DeadProc()
    respawned = []
    monkeypatch.setattr(driver.pool, "respawn",
                        lambda w: respawned.append(w.worker_id),
                        raising=False)
    driver._reap_dead()
    assert respawned == [0]
    trial = driver.get_trial(lost)
    assert trial.status == Trial.SCHEDULED
    assert trial.metric_history == []
    # re-registration gets the SAME trial back (BLACK semantics)
    w0.trial_id = None
    w0.process = None
    assert driver._assign_pending(w0)
    assert w0.trial_id == lost


def test_unknown_final_is_tolerated(driver):
    w0 = reg(driver, 0)
    before = w0.trial_id
    driver._handle_final(
        w0, (M.FINAL, 0, "ffffffffffffffff", 1.0, 0.1, False, ""))
    # unknown trial logged; worker got a fresh assignment
    assert w0.trial_id is not None
    assert driver.get_trial(before) is not None


def test_error_marks_trial_and_continues(driver):
    w0 = reg(driver, 0)
    tid = w0.trial_id
    driver._handle_error(w0, (M.ERROR, 0, tid, "traceback"))
    assert driver._error_store[0].trial_id == tid
    assert driver._error_store[0].status == Trial.ERROR
    assert w0.trial_id is not None and w0.trial_id != tid
