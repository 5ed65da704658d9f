"""Auxiliary subsystems: tensorboard shim, callbacks, checkpointing."""
import json
import os

import torch

from maggy_amd import tensorboard
from maggy_amd.callbacks import BatchEnd, EpochEnd, KerasEpochEnd
from maggy_amd.core.reporter import Reporter
from maggy_amd.models import MLP
from maggy_amd.utils.checkpoint import (
    load_checkpoint,
    load_finished_trials,
    save_checkpoint,
)


def test_tensorboard_registry(tmp_path):
    d = str(tmp_path / "trial_x")
    os.makedirs(d)
    tensorboard._register(d)
    assert tensorboard.logdir() == d
    tensorboard._write_hparams({"lr": 0.1, "act": "relu"}, "trial_x")
    summary = json.load(open(os.path.join(d, ".hparams_summary.json")))
    assert summary["hparams"]["lr"] == 0.1
    tensorboard.add_scalar("loss", 1.0, 0)  # no-op without tensorboard pkg
    tensorboard._reset()
    assert tensorboard.logdir() is None


def test_callbacks_report():
    rep = Reporter()
    BatchEnd(rep, "loss")(0, {"loss": 0.5})
    EpochEnd(rep, "val")(1, {"val": 0.7})
    cb = KerasEpochEnd(rep, "acc")
    cb.on_epoch_end(2, {"acc": 0.9})
    assert rep.metric == 0.9 and rep.step == 2


def test_checkpoint_roundtrip(tmp_path):
    d = str(tmp_path)
    model = MLP(in_features=8, hidden=4, num_classes=2)
    opt = torch.optim.SGD(model.parameters(), lr=0.1)
    save_checkpoint(d, model, opt, step=7, extra={"k": 1})
    model2 = MLP(in_features=8, hidden=4, num_classes=2)
    state = load_checkpoint(d, model2)
    assert state["step"] == 7 and state["extra"] == {"k": 1}
    for p, q in zip(model.parameters(), model2.parameters()):
        assert torch.equal(p, q)


def test_load_finished_trials(tmp_path):
    from maggy_amd import Trial

    t = Trial({"lr": 0.5})
    t.status = Trial.FINALIZED
    t.final_metric = 1.5
    os.makedirs(tmp_path / t.trial_id)
    with open(tmp_path / t.trial_id / "trial.json", "w") as f:
        f.write(t.to_json())
    loaded = load_finished_trials(str(tmp_path))
    assert len(loaded) == 1
    assert loaded[0].trial_id == t.trial_id
    assert loaded[0].final_metric == 1.5
