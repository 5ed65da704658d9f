"""TensorBoard surface: per-trial logdir registry + HParams records.

Parity: /root/reference/maggy/tensorboard.py:27-107 — the reference wires
the TF HParams plugin; this framework has no TF, so the same API writes
TensorBoard event files when ``torch.utils.tensorboard`` is importable
(requires the tensorboard package) and always writes the plain-JSON
``.hparams_summary.json`` fallback so the artifact surface exists either
way.  ``tensorboard.logdir()`` keeps its contract inside training
functions.
"""
import json
import os
import threading

_lock = threading.Lock()
_logdir = None
_writer = None
_writer_cls = "unprobed"  # probe once per process: ~2 s on a miss


def _try_writer(logdir):
    global _writer_cls
    if _writer_cls == "unprobed":
        try:
            from torch.utils.tensorboard import SummaryWriter

            _writer_cls = SummaryWriter
        except Exception:
            _writer_cls = None
    if _writer_cls is None:
        return None
    try:
        return _writer_cls(log_dir=logdir)
    except Exception:
        return None


def _register(trial_dir):
    """Called by the trial executor when a trial starts."""
    global _logdir, _writer
    with _lock:
        if _writer is not None:
            try:
                _writer.close()
            except Exception:
                pass
        _logdir = trial_dir
        _writer = _try_writer(trial_dir)


def logdir():
    """The current trial's log directory (user API)."""
    with _lock:
        return _logdir


def add_scalar(tag, value, step=None):
    """Log a scalar to the trial's TensorBoard (no-op without tensorboard;
    the metric still reaches the driver via reporter.broadcast)."""
    with _lock:
        if _writer is not None:
            _writer.add_scalar(tag, value, global_step=step)


def _write_hparams_config(searchspace, expected_metrics=("Metric",)):
    """Experiment-level hparams domain record (driver side)."""
    with _lock:
        if _logdir is None:
            return
        path = os.path.join(_logdir, ".hparams_config.json")
    with open(path, "w") as f:
        json.dump({
            "hparams": searchspace.to_dict() if searchspace else {},
            "metrics": list(expected_metrics),
        }, f)


def _write_hparams(hparams, trial_id):
    """Per-trial hparams record."""
    with _lock:
        ld = _logdir
        w = _writer
    if ld is not None:
        with open(os.path.join(ld, ".hparams_summary.json"), "w") as f:
            json.dump({"trial_id": trial_id, "hparams": hparams}, f)
    if w is not None:
        try:
            w.add_hparams(
                {k: v for k, v in hparams.items()
                 if isinstance(v, (int, float, str, bool))}, {})
        except Exception:
            pass


def _reset():
    global _logdir, _writer
    with _lock:
        if _writer is not None:
            try:
                _writer.close()
            except Exception:
                pass
        _writer = None
        _logdir = None
