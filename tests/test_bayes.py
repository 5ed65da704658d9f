"""GP / TPE optimizer tests with a deterministic fake experiment loop."""
import numpy as np
import pytest

from maggy_amd import Searchspace, Trial
from maggy_amd.optimizer import resolve_controller
from maggy_amd.optimizer.bayes import GP, TPE


class FakeDriver:
    def __init__(self, searchspace, num_trials, direction="min"):
        self.searchspace = searchspace
        self.num_trials = num_trials
        self.direction = direction
        self._trial_store = {}
        self._final_store = []


def run_sequential(opt, driver, objective, max_iters=200):
    """Drive the optimizer through a sequential experiment."""
    finished = None
    n = 0
    while True:
        t = opt.get_suggestion(finished)
        if t is None:
            break
        assert t != "IDLE"  # sequential: never idle without pruner
        n += 1
        assert n <= max_iters
        driver._trial_store[t.trial_id] = t
        t.status = Trial.FINALIZED
        params = {k: v for k, v in t.params.items() if k != "budget"}
        t.final_metric = objective(params)
        driver._final_store.append(t)
        del driver._trial_store[t.trial_id]
        finished = t
    return driver._final_store


def quadratic(params):
    return (params["x"] - 0.3) ** 2 + (params["y"] + 0.2) ** 2


@pytest.mark.parametrize("opt_cls", [GP, TPE])
def test_bo_optimizes_quadratic(opt_cls):
    np.random.seed(0)
    import random

    random.seed(0)
    sp = Searchspace(x=("DOUBLE", [-1.0, 1.0]), y=("DOUBLE", [-1.0, 1.0]))
    d = FakeDriver(sp, 40, direction="min")
    opt = resolve_controller(
        opt_cls(num_warmup_trials=10, random_fraction=0.1), d)
    opt._initialize()
    finals = run_sequential(opt, d, quadratic)
    assert len(finals) == 40
    best = min(t.final_metric for t in finals)
    # BO over 40 evals should comfortably reach near the optimum; pure
    # random baseline of 40 draws has E[min] ~ 0.07
    assert best < 0.1
    # the model produced non-random samples
    assert any(t.info_dict.get("sample_type") == "model" for t in finals)


def test_gp_direction_max():
    np.random.seed(1)
    sp = Searchspace(x=("DOUBLE", [0.0, 1.0]))
    d = FakeDriver(sp, 25, direction="max")
    opt = resolve_controller(GP(num_warmup_trials=8, random_fraction=0.1), d)
    opt._initialize()
    finals = run_sequential(opt, d, lambda p: -(p["x"] - 0.7) ** 2)
    best = max(t.final_metric for t in finals)
    assert best > -0.05


def test_gp_categorical_support():
    np.random.seed(2)
    sp = Searchspace(x=("DOUBLE", [0.0, 1.0]),
                     act=("CATEGORICAL", ["a", "b", "c"]))
    d = FakeDriver(sp, 30, direction="min")
    opt = resolve_controller(GP(num_warmup_trials=10), d)
    opt._initialize()
    finals = run_sequential(
        opt, d,
        lambda p: p["x"] ** 2 + {"a": 0.0, "b": 0.5, "c": 1.0}[p["act"]])
    assert len(finals) == 30
    assert all(t.params["act"] in ("a", "b", "c") for t in finals)


def test_registry_strings():
    sp = Searchspace(x=("DOUBLE", [0.0, 1.0]))
    d = FakeDriver(sp, 5)
    assert isinstance(resolve_controller("gp", d), GP)
    assert isinstance(resolve_controller("tpe", d), TPE)
    with pytest.raises(ValueError):
        resolve_controller("bogus", d)


def test_gp_interim_results():
    """GP with BOHB-style interim augmentation: heartbeat metrics at
    fractional progress feed the surrogate alongside finals."""
    np.random.seed(3)
    import random

    random.seed(3)
    sp = Searchspace(x=("DOUBLE", [0.0, 1.0]))
    d = FakeDriver(sp, 30, direction="min")
    opt = resolve_controller(
        GP(num_warmup_trials=8, random_fraction=0.1,
           interim_results=True, interim_results_interval=2), d)
    opt._initialize()
    finished = None
    n = 0
    while True:
        t = opt.get_suggestion(finished)
        if t is None:
            break
        n += 1
        assert n <= 60
        d._trial_store[t.trial_id] = t
        # simulate a metric curve converging to the quadratic objective
        final = (t.params["x"] - 0.4) ** 2
        for s in range(6):
            t.append_metric({"value": final + (6 - s) * 0.1, "step": s})
        t.status = Trial.FINALIZED
        t.final_metric = final
        d._final_store.append(t)
        del d._trial_store[t.trial_id]
        finished = t
    assert len(d._final_store) == 30
    best = min(t.final_metric for t in d._final_store)
    assert best < 0.05
    # surrogate trained on augmented rows: D+1 columns
    X, y, n_fin = opt.get_XY()
    assert X.shape[1] == 2
    assert X.shape[0] > n_fin  # interim rows present


def test_gp_incumbent_original_scale():
    """EI's incumbent must be the ORIGINAL-scale y minimum, not sklearn's
    normalized y_train_: with a shifted objective
    (metric ~ +1000) the acquisition must still optimize."""
    np.random.seed(4)
    import random

    random.seed(4)
    sp = Searchspace(x=("DOUBLE", [0.0, 1.0]))
    d = FakeDriver(sp, 30, direction="min")
    opt = resolve_controller(GP(num_warmup_trials=8, random_fraction=0.1), d)
    opt._initialize()
    finals = run_sequential(
        opt, d, lambda p: 1000.0 + (p["x"] - 0.6) ** 2)
    # incumbent bookkeeping is original scale
    assert abs(opt._y_fit_min[0] - min(t.final_metric
                                       for t in finals)) < 1e-9
    assert min(t.final_metric for t in finals) < 1000.005
    assert any(t.info_dict.get("sample_type") == "model" for t in finals)


def test_gp_kriging_believer():
    """Kriging-believer imputation: busy locations get the model's own
    posterior mean, not a constant (reference gp.py:329-373)."""
    np.random.seed(5)
    import random

    random.seed(5)
    sp = Searchspace(x=("DOUBLE", [0.0, 1.0]))
    d = FakeDriver(sp, 20, direction="min")
    opt = resolve_controller(
        GP(num_warmup_trials=6, random_fraction=0.1,
           imputed_metric="kb"), d)
    opt._initialize()
    # finish 6 warmup trials so a model exists
    finished = None
    for _ in range(6):
        t = opt.get_suggestion(finished)
        d._trial_store[t.trial_id] = t
        t.status = Trial.FINALIZED
        t.final_metric = (t.params["x"] - 0.5) ** 2
        d._final_store.append(t)
        del d._trial_store[t.trial_id]
        finished = t
    opt.update_model(0)
    # park a busy trial at x=0.5 (the optimum): KB should impute a value
    # near 0, far from cl_min-of-warmup only if warmup hit near 0.5
    busy = Trial({"x": 0.5})
    d._trial_store[busy.trial_id] = busy
    X, y, n_fin = opt.get_XY(include_busy=True)
    assert X.shape[0] == n_fin + 1
    kb_value = y[-1]
    model_mu = float(opt.models[0].predict(np.asarray([[0.5]]))[0])
    assert abs(kb_value - model_mu) < 1e-9


def test_gp_invalid_imputed_metric():
    with pytest.raises(ValueError):
        GP(imputed_metric="bogus")


def test_gp_lbfgs_refinement_improves_acquisition():
    """The L-BFGS-B polish never returns a candidate with worse acquisition
    than the best sampled point."""
    np.random.seed(6)
    sp = Searchspace(x=("DOUBLE", [0.0, 1.0]), y=("DOUBLE", [0.0, 1.0]))
    d = FakeDriver(sp, 10, direction="min")
    opt = resolve_controller(
        GP(num_warmup_trials=6, random_fraction=0.0, acq_n_points=50), d)
    opt._initialize()
    finished = None
    for _ in range(6):
        t = opt.get_suggestion(finished)
        d._trial_store[t.trial_id] = t
        t.status = Trial.FINALIZED
        t.final_metric = (t.params["x"] - 0.3) ** 2 + (t.params["y"]) ** 2
        d._final_store.append(t)
        del d._trial_store[t.trial_id]
        finished = t
    opt.update_model(0)
    model = opt.models[0]
    y_best = opt._y_fit_min[0]
    np.random.seed(7)
    cand = np.random.uniform(0, 1, size=(50, 2))
    mu, sigma = model.predict(cand, return_std=True)
    score = opt._acquisition(mu, sigma, y_best)
    order = np.argsort(score)[::-1]
    sampled_best = cand[int(order[0])]
    refined = opt._refine_lbfgs(model, y_best, cand[order[:5]], sampled_best)

    def acq(x):
        m, s = model.predict(x.reshape(1, -1), return_std=True)
        return float(opt._acquisition(m, s, y_best)[0])

    assert acq(refined) >= acq(sampled_best) - 1e-12
