from maggy_amd import Trial


def test_trial_init_reference_hash_vector():
    """The content-addressed id must match the reference implementation
    (reference test vector: maggy/tests/test_trial.py:33)."""
    trial = Trial({"param1": 5, "param2": "ada"})
    assert trial.params == {"param1": 5, "param2": "ada"}
    assert trial.status == Trial.PENDING
    assert trial.trial_id == "3d1cc9fdb1d4d001"


def test_trial_serialization_roundtrip():
    trial = Trial({"param1": 5, "param2": "ada"})
    trial.append_metric({"value": 1.5, "step": 0})
    trial.append_metric({"value": 2.5, "step": 1})
    new = Trial.from_json(trial.to_json())
    assert isinstance(new, Trial)
    assert new.trial_id == "3d1cc9fdb1d4d001"
    assert new.metric_history == [1.5, 2.5]
    assert new.status == Trial.PENDING


def test_append_metric_dedup_by_step():
    t = Trial({"a": 1})
    assert t.append_metric({"value": 1.0, "step": 0}) == 0
    assert t.append_metric({"value": 9.0, "step": 0}) is None  # dup step
    assert t.append_metric({"value": 2.0, "step": 1}) == 1
    assert t.metric_history == [1.0, 2.0]
    assert t.metric_dict == {0: 1.0, 1: 2.0}


def test_bad_params():
    import pytest

    with pytest.raises(ValueError):
        Trial("notadict")
    with pytest.raises(ValueError):
        Trial({1: "x"})


def test_early_stop_flag():
    t = Trial({"a": 1})
    assert not t.get_early_stop()
    t.set_early_stop()
    assert t.get_early_stop()
