"""Framework utilities: return-value handling, kwargs injection, device count.

Parity: /root/reference/maggy/util.py:159-199 (handle_return_val writes
.outputs.json + .metric per trial dir), util.py:79-94 (progress bar),
trial_executor.py:166-179 (signature-based kwargs injection).
"""
import inspect
import json
import os
import shutil

from maggy_amd import constants, exceptions
from maggy_amd.utils.jsonutil import json_default_numpy


def num_gpus():
    """Number of visible GPUs (0 on a CPU-only host)."""
    try:
        import torch

        if torch.cuda.is_available():
            return torch.cuda.device_count()
    except Exception:
        pass
    return 0


def handle_return_val(return_val, log_dir, optimization_key, log_file=None):
    """Validate the training-function return value and persist the per-trial
    artifacts ``.outputs.json`` and ``.metric``.

    Accepts a numeric scalar (wrapped as ``{optimization_key: value}``) or a
    dict that must contain ``optimization_key`` with a numeric value.
    Returns the scalar optimization value.
    """
    if not optimization_key:
        raise ValueError("Optimization key cannot be None.")
    if return_val is None or not isinstance(
        return_val, constants.USER_FCT.RETURN_TYPES
    ):
        raise exceptions.ReturnTypeError(optimization_key, return_val)
    if isinstance(return_val, dict):
        if optimization_key not in return_val:
            raise KeyError(
                "Returned dictionary does not contain optimization key with "
                "the provided name: {}".format(optimization_key)
            )
        opt_val = return_val[optimization_key]
    else:
        opt_val = return_val
        return_val = {optimization_key: opt_val}
    if not isinstance(opt_val, constants.USER_FCT.NUMERIC_TYPES):
        raise exceptions.MetricTypeError(optimization_key, opt_val)

    if log_file is not None:
        return_val = dict(return_val)
        return_val["log"] = log_file

    os.makedirs(log_dir, exist_ok=True)
    with open(os.path.join(log_dir, ".outputs.json"), "w") as f:
        f.write(json.dumps(return_val, default=json_default_numpy))
    with open(os.path.join(log_dir, ".metric"), "w") as f:
        f.write(json.dumps(opt_val, default=json_default_numpy))
    return opt_val


def build_train_kwargs(train_fn, model=None, dataset=None, hparams=None,
                       reporter=None, extra=None):
    """Inject only the kwargs the user function declares (parity:
    trial_executor.py:166-179)."""
    sig = inspect.signature(train_fn)
    kwargs = {}
    candidates = {
        "model": model,
        "dataset": dataset,
        "hparams": hparams,
        "reporter": reporter,
    }
    if extra:
        candidates.update(extra)
    for name, value in candidates.items():
        if name in sig.parameters:
            kwargs[name] = value
    return kwargs


def clean_dir(path, keep=()):
    """Remove everything inside ``path`` except the paths in ``keep``."""
    keep = {os.path.abspath(k) for k in keep}
    for entry in os.listdir(path):
        p = os.path.join(path, entry)
        if os.path.abspath(p) in keep:
            continue
        if os.path.isdir(p):
            shutil.rmtree(p, ignore_errors=True)
        else:
            try:
                os.remove(p)
            except OSError:
                pass


def progress_bar(done, total, width=24):
    frac = 0 if total == 0 else done / total
    filled = int(width * frac)
    return "[{}{}] {}/{}".format("#" * filled, "-" * (width - filled), done, total)


def seconds_to_milliseconds(seconds):
    return int(round(seconds * 1000))
