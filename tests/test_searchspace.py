import numpy as np
import pytest

from maggy_amd import Searchspace


def test_init_and_types():
    sp = Searchspace(
        lr=("DOUBLE", [0.01, 0.1]),
        layers=("INTEGER", [1, 5]),
        bs=("DISCRETE", [32, 64, 128]),
        act=("CATEGORICAL", ["relu", "gelu"]),
    )
    assert sp.names() == {
        "lr": "DOUBLE", "layers": "INTEGER", "bs": "DISCRETE",
        "act": "CATEGORICAL",
    }
    assert sp.get("lr") == [0.01, 0.1]
    assert sp.lr == [0.01, 0.1]
    assert "lr" in sp
    assert sp.get("nope", 42) == 42


def test_add_validation_errors():
    sp = Searchspace()
    with pytest.raises(ValueError):
        sp.add("x", ("DOUBLE", []))
    with pytest.raises(ValueError):
        sp.add("x", ("DOUBLE", [1.0]))
    with pytest.raises(ValueError):
        sp.add("x", ("WEIRD", [1, 2]))
    with pytest.raises(ValueError):
        sp.add("x", ("DOUBLE", [2.0, 1.0]))
    with pytest.raises(ValueError):
        sp.add("x", ("INTEGER", [0.5, 1.5]))
    with pytest.raises(ValueError):
        sp.add("x", "notatuple")
    sp.add("x", ("DOUBLE", [0.0, 1.0]))
    with pytest.raises(ValueError):
        sp.add("x", ("DOUBLE", [0.0, 1.0]))  # reserved


def test_random_sampling_bounds():
    sp = Searchspace(
        lr=("DOUBLE", [0.01, 0.1]),
        n=("INTEGER", [1, 5]),
        act=("CATEGORICAL", ["a", "b"]),
    )
    for params in sp.get_random_parameter_values(50):
        assert 0.01 <= params["lr"] <= 0.1
        assert 1 <= params["n"] <= 5 and isinstance(params["n"], int)
        assert params["act"] in ("a", "b")


def test_transform_roundtrip():
    sp = Searchspace(
        lr=("DOUBLE", [0.01, 0.1]),
        n=("INTEGER", [1, 5]),
        act=("CATEGORICAL", ["a", "b", "c"]),
    )
    cfg = {"lr": 0.055, "n": 3, "act": "b"}
    as_list = sp.dict_to_list(cfg)
    t = sp.transform(as_list, normalize_categorical=True)
    assert all(0.0 <= v <= 1.0 for v in t)
    back = sp.inverse_transform(t, normalize_categorical=True)
    restored = sp.list_to_dict(back)
    assert restored["n"] == 3
    assert restored["act"] == "b"
    assert abs(restored["lr"] - 0.055) < 1e-9


def test_deterministic_rng_sample():
    sp = Searchspace(lr=("DOUBLE", [0.0, 1.0]), n=("INTEGER", [0, 9]))
    a = sp.sample(np.random.default_rng(0))
    b = sp.sample(np.random.default_rng(0))
    assert a == b
