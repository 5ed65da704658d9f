"""Hyperband resume: bracket reconstruction from persisted trials."""
import random

from maggy_amd import Searchspace, Trial
from maggy_amd.optimizer import RandomSearch, resolve_controller


class FakeDriver:
    def __init__(self, searchspace, num_trials, direction="min"):
        self.searchspace = searchspace
        self.num_trials = num_trials
        self.direction = direction
        self._trial_store = {}
        self._final_store = []


def make_opt(driver):
    return resolve_controller(
        RandomSearch(pruner="hyperband",
                     pruner_kwargs=dict(min_budget=1, max_budget=4, eta=2,
                                        n_iterations=1)), driver)


def run_until(opt, driver, stop_after=None):
    finished = None
    done = []
    while True:
        t = opt.get_suggestion(finished)
        if t is None or t == "IDLE":
            return done, t
        driver._trial_store[t.trial_id] = t
        t.status = Trial.FINALIZED
        t.final_metric = t.params["lr"]
        driver._final_store.append(t)
        del driver._trial_store[t.trial_id]
        done.append(t)
        finished = t
        if stop_after is not None and len(done) >= stop_after:
            return done, None


def roundtrip(trials):
    """Persist + reload like the driver's resume path does."""
    return [Trial.from_json(t.to_json()) for t in trials]


def test_hyperband_resume_partial():
    random.seed(5)
    sp = Searchspace(lr=("DOUBLE", [0.0, 1.0]))
    # bracket [4, 2, 1] at budgets [1, 2, 4] = 7 trials total; stop after 5
    d1 = FakeDriver(sp, 100)
    opt1 = make_opt(d1)
    opt1._initialize()
    done, _ = run_until(opt1, d1, stop_after=5)
    assert len(done) == 5

    # resume into a fresh controller
    d2 = FakeDriver(sp, 100)
    d2._final_store.extend(roundtrip(done))
    opt2 = make_opt(d2)
    opt2._initialize()
    opt2.on_resume(d2._final_store)
    more, _ = run_until(opt2, d2)
    # exactly the remaining 2 trials run; budget multiset over both halves
    # matches a fresh full bracket
    assert len(more) == 2
    budgets = sorted(int(t.params["budget"]) for t in d2._final_store)
    assert budgets == [1, 1, 1, 1, 2, 2, 4]
    assert opt2.pruner.finished()


def test_hyperband_resume_complete():
    random.seed(6)
    sp = Searchspace(lr=("DOUBLE", [0.0, 1.0]))
    d1 = FakeDriver(sp, 100)
    opt1 = make_opt(d1)
    opt1._initialize()
    done, _ = run_until(opt1, d1)
    assert len(done) == 7

    d2 = FakeDriver(sp, 100)
    d2._final_store.extend(roundtrip(done))
    opt2 = make_opt(d2)
    opt2._initialize()
    opt2.on_resume(d2._final_store)
    assert opt2.pruner.finished()
    assert opt2.get_suggestion() is None
