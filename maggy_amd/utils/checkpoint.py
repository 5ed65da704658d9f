"""Per-trial model checkpointing.

The reference persists only metrics/logs (Trial.from_json
exists but resume is latent).  With node-local storage this framework adds
optional model checkpoints in the trial directory and experiment resume
from the persisted trial.json records.
"""
import os

import torch

from maggy_amd.trial import Trial


def save_checkpoint(trial_dir, model, optimizer=None, step=None, extra=None):
    """Write ``checkpoint.pt`` into the trial dir (atomic rename)."""
    state = {
        "model": model.state_dict(),
        "optimizer": optimizer.state_dict() if optimizer is not None else None,
        "step": step,
        "extra": extra,
    }
    tmp = os.path.join(trial_dir, ".checkpoint.pt.tmp")
    dst = os.path.join(trial_dir, "checkpoint.pt")
    torch.save(state, tmp)
    os.replace(tmp, dst)
    return dst


def load_checkpoint(trial_dir, model=None, optimizer=None,
                    map_location="cpu"):
    """Load ``checkpoint.pt``; returns the raw state dict (and loads into
    model/optimizer when given)."""
    path = os.path.join(trial_dir, "checkpoint.pt")
    state = torch.load(path, map_location=map_location, weights_only=False)
    if model is not None:
        model.load_state_dict(state["model"])
    if optimizer is not None and state.get("optimizer"):
        optimizer.load_state_dict(state["optimizer"])
    return state


def load_finished_trials(exp_dir):
    """Recover finalized Trial objects from an experiment directory (the
    resume path the reference left latent: trial.py from_json)."""
    trials = []
    for entry in sorted(os.listdir(exp_dir)):
        tj = os.path.join(exp_dir, entry, "trial.json")
        if os.path.isfile(tj):
            with open(tj) as f:
                trials.append(Trial.from_json(f.read()))
    return trials
