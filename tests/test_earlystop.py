from maggy_amd import Trial
from maggy_amd.earlystop import MedianStoppingRule, NoStoppingRule


def make_final(history):
    t = Trial({"h": str(history)})
    t.status = Trial.FINALIZED
    t.metric_history = list(history)
    t.final_metric = history[-1]
    return t


def test_median_rule_stops_underperformer_max():
    finalized = [make_final([10, 20, 30]), make_final([20, 30, 40])]
    bad = Trial({"x": 1})
    bad.metric_history = [1, 2]
    # medians of running averages at step 2: [(10+20)/2, (20+30)/2] -> 20
    assert MedianStoppingRule.earlystop_check(bad, finalized, "max") == \
        bad.trial_id
    good = Trial({"x": 2})
    good.metric_history = [50, 60]
    assert MedianStoppingRule.earlystop_check(good, finalized, "max") is None


def test_median_rule_min_direction():
    finalized = [make_final([10, 20]), make_final([20, 30])]
    bad = Trial({"x": 1})
    bad.metric_history = [100, 200]
    assert MedianStoppingRule.earlystop_check(bad, finalized, "min") == \
        bad.trial_id
    good = Trial({"x": 2})
    good.metric_history = [1, 2]
    assert MedianStoppingRule.earlystop_check(good, finalized, "min") is None


def test_median_rule_no_history():
    t = Trial({"x": 1})
    assert MedianStoppingRule.earlystop_check(t, [], "max") is None


def test_nostop():
    t = Trial({"x": 1})
    t.metric_history = [0.0]
    assert NoStoppingRule.earlystop_check(t, [], "max") is None
