"""Optimizer unit tests with a fake driver (no pool, no GPU)."""
import pytest

from maggy_amd import Searchspace, Trial
from maggy_amd.optimizer import (
    Asha,
    GridSearch,
    RandomSearch,
    SingleRun,
    resolve_controller,
)


class FakeDriver:
    def __init__(self, searchspace, num_trials, direction="max"):
        self.searchspace = searchspace
        self.num_trials = num_trials
        self.direction = direction
        self._trial_store = {}
        self._final_store = []


def finalize(controller, trial, metric, driver):
    trial.status = Trial.FINALIZED
    trial.final_metric = metric
    driver._final_store.append(trial)
    driver._trial_store.pop(trial.trial_id, None)


def test_randomsearch_buffer():
    sp = Searchspace(lr=("DOUBLE", [0.0, 1.0]))
    d = FakeDriver(sp, 5)
    opt = resolve_controller("randomsearch", d)
    opt._initialize()
    trials = []
    t = opt.get_suggestion()
    while t is not None and t != "IDLE":
        trials.append(t)
        d._trial_store[t.trial_id] = t
        t = opt.get_suggestion(t)
    assert len(trials) == 5
    assert all(0.0 <= tr.params["lr"] <= 1.0 for tr in trials)
    assert opt.get_suggestion() is None


def test_gridsearch_cartesian():
    sp = Searchspace(a=("DISCRETE", [1, 2, 3]), b=("CATEGORICAL", ["x", "y"]))
    d = FakeDriver(sp, 999)
    opt = resolve_controller("gridsearch", d)
    opt._initialize()
    assert opt.num_trials == 6
    seen = set()
    t = opt.get_suggestion()
    while t is not None:
        seen.add((t.params["a"], t.params["b"]))
        t = opt.get_suggestion(t)
    assert seen == {(a, b) for a in (1, 2, 3) for b in ("x", "y")}


def test_gridsearch_rejects_continuous():
    sp = Searchspace(lr=("DOUBLE", [0.0, 1.0]))
    d = FakeDriver(sp, 10)
    opt = resolve_controller("gridsearch", d)
    with pytest.raises(NotImplementedError):
        opt._initialize()


def test_singlerun():
    d = FakeDriver(None, 3)
    opt = resolve_controller(None, d)
    opt._initialize()
    ids = set()
    t = opt.get_suggestion()
    while t is not None:
        ids.add(t.trial_id)
        t = opt.get_suggestion(t)
    assert len(ids) == 3


def _run_asha(direction):
    sp = Searchspace(lr=("DOUBLE", [0.0, 1.0]))
    d = FakeDriver(sp, 16, direction=direction)
    opt = resolve_controller(Asha(reduction_factor=2, resource_min=1,
                                  resource_max=4), d)
    opt._initialize()
    # simulate a sequential experiment: metric = lr (higher lr "better" for
    # max); run until the controller signals done
    finished = None
    budgets = []
    n = 0
    while True:
        t = opt.get_suggestion(finished)
        if t is None:
            break
        assert t != "IDLE"
        n += 1
        assert n < 200
        budgets.append(t.params["budget"])
        d._trial_store[t.trial_id] = t
        metric = t.params["lr"] if direction == "max" else -t.params["lr"]
        finalize(opt, t, metric, d)
        finished = t
    # rung structure: max_rung = log2(4) = 2; experiment ends when a trial
    # lands in rung 2 (budget 4)
    assert opt.max_rung == 2
    assert 2 in opt.rungs and len(opt.rungs[2]) >= 1
    assert set(budgets) == {1, 2, 4}
    # promotions must be the best of their rung at promotion time
    for promoted_id in opt.promoted[0]:
        src = next(t for t in opt.rungs[0] if t.trial_id == promoted_id)
        assert src.final_metric is not None
    return opt


def test_asha_promotion_max():
    _run_asha("max")


def test_asha_promotion_min():
    _run_asha("min")


def test_asha_validation():
    with pytest.raises(ValueError):
        Asha(reduction_factor=1)
    with pytest.raises(ValueError):
        Asha(resource_min=4, resource_max=2)
    sp = Searchspace(lr=("DOUBLE", [0.0, 1.0]))
    d = FakeDriver(sp, 2)  # too few trials for the rung structure
    opt = resolve_controller(Asha(2, 1, 4), d)
    with pytest.raises(ValueError):
        opt._initialize()


def test_duplicate_detection():
    sp = Searchspace(a=("DISCRETE", [1, 2]))
    d = FakeDriver(sp, 4)
    opt = resolve_controller("randomsearch", d)
    opt._initialize()
    t1 = Trial({"a": 1})
    d._trial_store[t1.trial_id] = t1
    dup = Trial({"a": 1, "budget": 3})  # same config, different budget
    assert opt.hparams_exist(dup)
    assert not opt.hparams_exist(Trial({"a": 2}))


def test_min_convention_metric_getters():
    sp = Searchspace(a=("DISCRETE", [1, 2]))
    d = FakeDriver(sp, 4, direction="max")
    opt = resolve_controller("randomsearch", d)
    for metric in (1.0, 3.0, 2.0):
        t = Trial({"a": metric})
        t.status = Trial.FINALIZED
        t.final_metric = metric
        d._final_store.append(t)
    # max experiment -> internal minimization convention negates
    assert opt.ybest() == -3.0
    assert opt.yworst() == -1.0
    assert opt.ymean() == -2.0
