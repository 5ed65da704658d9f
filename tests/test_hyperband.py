"""Hyperband pruner driven through RandomSearch in a simulated experiment."""
import random

import numpy as np
import pytest

from maggy_amd import Searchspace, Trial
from maggy_amd.optimizer import RandomSearch, resolve_controller
from maggy_amd.pruner import Hyperband


class FakeDriver:
    def __init__(self, searchspace, num_trials, direction="min"):
        self.searchspace = searchspace
        self.num_trials = num_trials
        self.direction = direction
        self._trial_store = {}
        self._final_store = []


def test_hyperband_bracket_structure():
    hb = Hyperband(min_budget=1, max_budget=9, eta=3, n_iterations=2,
                   trial_metric_getter=lambda ids: {})
    assert hb.max_rungs == 3
    assert hb.budget_ladder == [1, 3, 9]
    assert hb.brackets[0].n_configs == [9, 3, 1]
    assert hb.brackets[0].budgets == [1, 3, 9]
    assert hb.brackets[1].n_configs == [3, 1]
    assert hb.brackets[1].budgets == [3, 9]
    assert hb.num_trials() == 17


def test_hyperband_validation():
    with pytest.raises(ValueError):
        Hyperband(min_budget=0, max_budget=9, trial_metric_getter=None)
    with pytest.raises(ValueError):
        Hyperband(min_budget=9, max_budget=9, trial_metric_getter=None)
    with pytest.raises(ValueError):
        Hyperband(min_budget=1, max_budget=9, eta=1,
                  trial_metric_getter=None)


def test_randomsearch_with_hyperband_sequential():
    random.seed(0)
    np.random.seed(0)
    sp = Searchspace(lr=("DOUBLE", [0.0, 1.0]))
    d = FakeDriver(sp, 100, direction="min")
    opt = resolve_controller(
        RandomSearch(pruner="hyperband",
                     pruner_kwargs=dict(min_budget=1, max_budget=9, eta=3,
                                        n_iterations=2)), d)
    opt._initialize()
    finished = None
    budgets_run = []
    n = 0
    while True:
        t = opt.get_suggestion(finished)
        if t is None:
            break
        assert t != "IDLE"
        n += 1
        assert n <= 30
        budgets_run.append(t.params["budget"])
        d._trial_store[t.trial_id] = t
        t.status = Trial.FINALIZED
        t.final_metric = t.params["lr"]  # lower lr is better
        d._final_store.append(t)
        del d._trial_store[t.trial_id]
        finished = t
    assert n == 17
    # bracket 0: 9x budget1, 3x budget3, 1x budget9;
    # bracket 1: 3x budget3, 1x budget9
    assert budgets_run.count(1) == 9
    assert budgets_run.count(3) == 6
    assert budgets_run.count(9) == 2
    assert opt.pruner.finished()
    # promotions carry the best (lowest lr) configs forward
    b0 = opt.pruner.brackets[0]
    rung0_metrics = {s["actual"]: None for s in b0.slots[0]}
    lrs = {t.trial_id: t.params["lr"] for t in d._final_store}
    promoted_lr = [lrs[s["original"]] for s in b0.slots[1]]
    all_rung0 = sorted(lrs[tid] for tid in rung0_metrics)
    assert sorted(promoted_lr) == all_rung0[:3]


def test_hyperband_idle_when_busy():
    """Async: with unfinished trials in flight, the pruner goes IDLE rather
    than over-scheduling."""
    random.seed(1)
    sp = Searchspace(lr=("DOUBLE", [0.0, 1.0]))
    d = FakeDriver(sp, 100, direction="min")
    opt = resolve_controller(
        RandomSearch(pruner="hyperband",
                     pruner_kwargs=dict(min_budget=1, max_budget=4, eta=2,
                                        n_iterations=1)), d)
    opt._initialize()
    # bracket 0: n_configs [4,2,1] budgets [1,2,4]
    started = []
    for _ in range(4):
        t = opt.get_suggestion()
        assert isinstance(t, Trial)
        d._trial_store[t.trial_id] = t
        started.append(t)
    # all 4 rung-0 slots handed out, none finished -> IDLE
    assert opt.get_suggestion() == "IDLE"
    # finish them; promotions should flow
    for t in started:
        t.status = Trial.FINALIZED
        t.final_metric = t.params["lr"]
        d._final_store.append(t)
        del d._trial_store[t.trial_id]
    t = opt.get_suggestion(started[-1])
    assert isinstance(t, Trial)
    assert t.params["budget"] == 2


def _finish(d, t, metric):
    t.status = Trial.FINALIZED
    t.final_metric = metric
    d._final_store.append(t)
    d._trial_store.pop(t.trial_id, None)


def test_hyperband_errored_trial_is_rerun():
    """An errored trial frees its slot for a re-run instead of stalling the
    bracket."""
    random.seed(2)
    np.random.seed(2)
    sp = Searchspace(lr=("DOUBLE", [0.0, 1.0]))
    d = FakeDriver(sp, 100, direction="min")
    opt = resolve_controller(
        RandomSearch(pruner="hyperband",
                     pruner_kwargs=dict(min_budget=1, max_budget=4, eta=2,
                                        n_iterations=1)), d)
    opt._initialize()
    # bracket 0: n_configs [4, 2, 1], budgets [1, 2, 4]
    first = opt.get_suggestion()
    d._trial_store[first.trial_id] = first
    # the first rung-0 trial errors
    first.status = Trial.ERROR
    d._trial_store.pop(first.trial_id, None)
    opt.on_trial_error(first)
    # the bracket must still complete: 7 successful trials total
    finished = None
    n = 0
    budgets_run = []
    while True:
        t = opt.get_suggestion(finished)
        if t is None:
            break
        assert t != "IDLE", "bracket stalled after an errored trial"
        n += 1
        assert n <= 20
        budgets_run.append(t.params["budget"])
        d._trial_store[t.trial_id] = t
        _finish(d, t, t.params["lr"])
        finished = t
    assert n == 7
    assert budgets_run.count(1) == 4
    assert opt.pruner.finished()


def test_hyperband_error_in_promoted_rung():
    """An error in a promoted (rung>0) trial re-runs the same parent."""
    random.seed(3)
    np.random.seed(3)
    sp = Searchspace(lr=("DOUBLE", [0.0, 1.0]))
    d = FakeDriver(sp, 100, direction="min")
    opt = resolve_controller(
        RandomSearch(pruner="hyperband",
                     pruner_kwargs=dict(min_budget=1, max_budget=4, eta=2,
                                        n_iterations=1)), d)
    opt._initialize()
    finished = None
    for _ in range(4):  # complete rung 0
        t = opt.get_suggestion(finished)
        d._trial_store[t.trial_id] = t
        _finish(d, t, t.params["lr"])
        finished = t
    promo = opt.get_suggestion(finished)
    assert promo.params["budget"] == 2
    parent = promo.info_dict["parent_trial_id"]
    promo.status = Trial.ERROR
    opt.on_trial_error(promo)
    # same parent must be re-handed
    retry = opt.get_suggestion()
    assert retry.params["budget"] == 2
    assert retry.info_dict["parent_trial_id"] == parent
    d._trial_store[retry.trial_id] = retry
    _finish(d, retry, retry.params["lr"])
    finished = retry
    n = 1
    while True:
        t = opt.get_suggestion(finished)
        if t is None:
            break
        assert t != "IDLE"
        n += 1
        assert n <= 20
        d._trial_store[t.trial_id] = t
        _finish(d, t, t.params["lr"])
        finished = t
    assert opt.pruner.finished()


def test_hyperband_retry_budget_exhausted_marks_failed():
    """When the bracket's retry budget is spent, errored slots count as
    finished-with-worst-case so rungs still complete (never hang)."""
    random.seed(4)
    np.random.seed(4)
    sp = Searchspace(lr=("DOUBLE", [0.0, 1.0]))
    d = FakeDriver(sp, 100, direction="min")
    opt = resolve_controller(
        RandomSearch(pruner="hyperband",
                     pruner_kwargs=dict(min_budget=1, max_budget=4, eta=2,
                                        n_iterations=1)), d)
    opt._initialize()
    br = opt.pruner.brackets[0]
    # every trial errors: retry budget = sum(n_configs) = 7, after which
    # slots are marked failed
    n_err = 0
    while not opt.pruner.finished():
        t = opt.get_suggestion()
        if t is None:
            break
        assert t != "IDLE", "stalled with all-erroring train_fn"
        d._trial_store[t.trial_id] = t
        t.status = Trial.ERROR
        d._trial_store.pop(t.trial_id, None)
        opt.on_trial_error(t)
        n_err += 1
        assert n_err <= 40
    # rung 0 completed via failed slots; nothing promotable -> FINISHED
    assert br.state == "FINISHED"
    assert opt.get_suggestion() is None
