import json
import os

import pytest
import torch.nn as nn

from maggy_amd import experiment
from maggy_amd.ablation import AblationStudy, LOCO, drop_layers
from maggy_amd.config import AblationConfig
from tests import _ablation_fns as fns


def test_ablationstudy_api():
    study = AblationStudy()
    study.features.include("f1", ["f2", "f3"])
    study.features.exclude("f3")
    assert study.features.list_all() == ["f1", "f2"]
    study.model.layers.include("blocks.0", "blocks.1")
    study.model.layers.include_groups(["blocks.0", "blocks.1"])
    study.model.layers.include_groups(prefix="blocks")
    assert study.model.layers.list_all() == ["blocks.0", "blocks.1"]
    assert len(study.model.layers.included_groups) == 2
    with pytest.raises(ValueError):
        study.model.layers.include_groups(["only_one"])
    with pytest.raises(ValueError):
        study.features.include(42)


def test_loco_trial_buffer():
    study = AblationStudy()
    study.features.include("a", "b")
    study.model.layers.include("l1")
    study.model.add_custom_generator("variant", lambda: None)
    loco = LOCO(study)
    assert loco.get_number_of_trials() == 5  # base + 2 feat + 1 layer + 1 custom
    loco.initialize()
    trials = []
    t = loco.get_trial()
    while t is not None:
        trials.append(t.params)
        t = loco.get_trial()
    assert {"ablated_feature": "None", "ablated_layer": "None"} in trials
    assert {"ablated_feature": "a", "ablated_layer": "None"} in trials
    assert {"ablated_feature": "None", "ablated_layer": "l1"} in trials
    assert {"ablated_feature": "None",
            "ablated_layer": "custom:variant"} in trials


def test_drop_layers():
    m = nn.Sequential()
    m.add_module("a", nn.Linear(4, 4))
    m.add_module("b", nn.Linear(4, 4))
    m2 = drop_layers(m, "a")
    assert isinstance(m2.a, nn.Identity)
    assert isinstance(m2.b, nn.Linear)
    with pytest.raises(ValueError):
        drop_layers(m, "zzz")


def test_drop_layers_group_and_prefix():
    from maggy_amd.models import SmallTransformer

    m = SmallTransformer(vocab_size=10, dim=8, n_heads=2, n_layers=3,
                         max_seq_len=4)
    drop_layers(m, "blocks.0+blocks.2")
    assert isinstance(m.blocks[0], nn.Identity)
    assert isinstance(m.blocks[2], nn.Identity)
    assert not isinstance(m.blocks[1], nn.Identity)
    m2 = SmallTransformer(vocab_size=10, dim=8, n_heads=2, n_layers=2,
                          max_seq_len=4)
    drop_layers(m2, "blocks*")
    assert all(isinstance(b, nn.Identity) for b in m2.blocks)


def test_loco_e2e(exp_dir):
    study = AblationStudy(model_generator=fns.model_gen,
                          dataset_generator=fns.dataset_gen)
    study.features.include("tail")
    study.model.layers.include("blocks.0", "blocks.1")
    study.model.layers.include_groups(["blocks.0", "blocks.1"])
    cfg = AblationConfig(study, ablator="loco", direction="max",
                         name="loco-e2e", num_workers=2)
    res = experiment.lagom(fns.ablation_train_fn, cfg)
    # base + 1 feature + 2 layers + 1 group = 5 trials
    assert res["num_trials"] == 5
    app = [d for d in os.listdir(exp_dir) if d.startswith("application_")][0]
    run_dir = os.path.join(exp_dir, app,
                           sorted(os.listdir(os.path.join(exp_dir, app)))[0])
    ablated = []
    n_params = {}
    for td in os.listdir(run_dir):
        hp = os.path.join(run_dir, td, ".hparams.json")
        if os.path.isdir(os.path.join(run_dir, td)) and os.path.exists(hp):
            p = json.load(open(hp))
            ablated.append((p["ablated_feature"], p["ablated_layer"]))
            out = json.load(open(os.path.join(run_dir, td, ".outputs.json")))
            n_params[p["ablated_layer"]] = out["n_params"]
    assert ("None", "None") in ablated
    assert ("tail", "None") in ablated
    assert ("None", "blocks.0") in ablated
    assert ("None", "blocks.0+blocks.1") in ablated
    # ablating layers must actually shrink the model
    assert n_params["blocks.0"] < n_params["None"]
    assert n_params["blocks.0+blocks.1"] < n_params["blocks.0"]
