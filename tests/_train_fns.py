"""Top-level training functions for pool tests (must be picklable by spawn)."""
import os
import time


def quick_fn(hparams, reporter):
    acc = 0.0
    for step in range(4):
        acc += hparams.get("lr", 0.1)
        reporter.broadcast(acc, step)
    return acc


def metric_eq_lr(hparams, reporter):
    reporter.broadcast(hparams["lr"], 0)
    return hparams["lr"]


def budgeted_fn(hparams, reporter):
    """ASHA-style: runs `budget` steps, metric grows with lr."""
    budget = int(hparams.get("budget", 1))
    acc = 0.0
    for step in range(budget):
        acc = hparams["lr"] * (step + 1)
        reporter.broadcast(acc, step)
    return acc


def slow_fn_for_earlystop(hparams, reporter):
    """Reports the (constant) level for 60 steps; underperformers should be
    early-stopped long before finishing."""
    level = hparams["level"]
    for step in range(60):
        reporter.broadcast(level, step)
        time.sleep(0.02)
    return level


def returns_dict_fn(hparams, reporter):
    reporter.broadcast(0.5, 0)
    return {"Metric": 0.5, "aux": 1.0}


def crashing_fn(hparams, reporter):
    if hparams.get("boom", 0) > 0.5:
        raise RuntimeError("deliberate train_fn failure")
    reporter.broadcast(1.0, 0)
    return 1.0


def suicide_fn(hparams, reporter):
    """Kills the worker process once to exercise the respawn/BLACK path.

    A sentinel file under MAGGY_LOG_DIR persists across worker respawns so
    the re-assigned trial succeeds on the second attempt.
    """
    sentinel = os.path.join(os.environ["MAGGY_LOG_DIR"], "died_once")
    if hparams.get("die", 0) > 0.5 and not os.path.exists(sentinel):
        open(sentinel, "w").close()
        os._exit(13)
    reporter.broadcast(2.0, 0)
    return 2.0


def single_run_fn(model, dataset, hparams, reporter):
    reporter.broadcast(1.0, 0)
    reporter.broadcast(2.0, 1)
    return {"Metric": 3.0, "extra": 7}


def continuation_fn(hparams, reporter, trial_dir, parent_checkpoint):
    """Budget-trial that continues from the parent's checkpoint: the
    'model' is a cumulative step counter persisted as checkpoint.pt."""
    import torch

    steps_done = 0
    if parent_checkpoint is not None:
        steps_done = torch.load(parent_checkpoint,
                                weights_only=False)["model"]["steps"]
    budget = int(hparams.get("budget", 1))
    for s in range(budget):
        steps_done += 1
        reporter.broadcast(float(steps_done), s)

    class _M:
        def state_dict(self):
            return {"steps": steps_done}

    from maggy_amd.utils.checkpoint import save_checkpoint

    save_checkpoint(trial_dir, _M(), step=steps_done)
    return float(steps_done)


def fails_once_fn(hparams, reporter):
    """Budget-trial that raises on the first-ever invocation (sentinel file
    under MAGGY_LOG_DIR survives across trials) — exercises the Hyperband
    errored-slot re-run path."""
    sentinel = os.path.join(os.environ["MAGGY_LOG_DIR"], "failed_once")
    if not os.path.exists(sentinel):
        open(sentinel, "w").close()
        raise RuntimeError("deliberate first-trial failure")
    return budgeted_fn(hparams, reporter)


def noisy_quadratic_fn(hparams, reporter):
    """Deterministic objective for pool BO tests: peak at lr=0.06."""
    v = 1.0 - (hparams["lr"] - 0.06) ** 2 * 100.0
    reporter.broadcast(v, 0)
    return v


def chatty_slow_fn(hparams, reporter):
    """Prints every step for ~3 s: feeds the live log-stream test."""
    for step in range(30):
        print("chatty step", step)
        reporter.broadcast(float(step), step)
        time.sleep(0.1)
    return 1.0
