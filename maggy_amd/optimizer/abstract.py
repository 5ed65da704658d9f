"""Abstract optimizer (controller) interface.

Parity: /root/reference/maggy/optimizer/abstractoptimizer.py:36-443 — same
``get_suggestion(trial) -> Trial | "IDLE" | None`` contract, per-optimizer
log file, min-convention metric getters (negate when direction=="max"),
duplicate-config detection, trial creation with run_budget / sample_type
info and budget injection into hparams, and pruner wiring.
"""
import time
from abc import ABC, abstractmethod

import numpy as np

from maggy_amd.trial import Trial


class AbstractOptimizer(ABC):
    experiment_type = "optimization"

    def __init__(self, pruner=None, pruner_kwargs=None):
        self.searchspace = None
        self.num_trials = None
        self.trial_store = None   # dict trial_id -> running Trial
        self.final_store = None   # list of finalized Trials
        self.direction = None     # "max" | "min"
        self.pruner = None
        self.interim_results = False
        self._log_fd = None
        if pruner:
            self.init_pruner(pruner, pruner_kwargs or {})

    def init_pruner(self, pruner, pruner_kwargs):
        """Attach a pruner (parity abstractoptimizer.py:297-315); the metric
        getter hands it min-convention finalized metrics."""
        from maggy_amd.pruner import AbstractPruner, Hyperband

        if isinstance(pruner, AbstractPruner):
            self.pruner = pruner
            if self.pruner.trial_metric_getter is None:
                self.pruner.trial_metric_getter = self.get_metrics_dict
        elif pruner == "hyperband":
            self.pruner = Hyperband(
                trial_metric_getter=self.get_metrics_dict, **pruner_kwargs)
        else:
            raise ValueError(
                "Unknown pruner '{}'; expected 'hyperband' or an "
                "AbstractPruner".format(pruner))

    # -- wiring (called by the driver) ----------------------------------
    def _initialize(self, exp_dir=None):
        if exp_dir is not None:
            self._log_fd = open(exp_dir + "/optimizer.log", "w")
        if self.pruner is not None:
            self.pruner._initialize(exp_dir=exp_dir)
        self.initialize()

    def _finalize_experiment(self, trials):
        self.finalize_experiment(trials)
        if self.pruner is not None:
            self.pruner._close_log()
        if self._log_fd is not None:
            self._log_fd.close()
            self._log_fd = None

    def _log(self, msg):
        if self._log_fd is not None and not self._log_fd.closed:
            self._log_fd.write(
                "{}: {}\n".format(time.strftime("%H:%M:%S"), msg))
            self._log_fd.flush()

    def name(self):
        return str(self.__class__.__name__)

    # -- the contract ----------------------------------------------------
    @abstractmethod
    def initialize(self):
        """Called once before the experiment starts."""

    @abstractmethod
    def get_suggestion(self, trial=None):
        """Return the next Trial, "IDLE" if none is ready yet, or None when
        the experiment is done. ``trial`` is the trial that just finished
        (None on first assignment)."""

    @abstractmethod
    def finalize_experiment(self, trials):
        """Called once after the last trial finalized."""

    def on_trial_error(self, trial):
        """Driver callback when an assigned trial errors (user exception or
        worker crash give-up).  Default: notify the pruner so its bracket
        slot is reset/failed instead of stalling the rung (an errored trial
        otherwise never reaches final_store and Hyperband's
        ``_rung_complete`` stays false forever).  Optimizers without budget
        state need no action — the driver simply asks for the next
        suggestion."""
        if self.pruner is not None and hasattr(self.pruner, "on_trial_error"):
            self.pruner.on_trial_error(trial.trial_id)

    # -- helpers shared by concrete optimizers ---------------------------
    def create_trial(self, hparams, sample_type="random", run_budget=0,
                     model_budget=None, parent_trial_id=None):
        """Build a Trial, stamping scheduling metadata and injecting the
        training budget into the hparams when budget-based (parity:
        abstractoptimizer.py:317-376).  ``parent_trial_id`` marks promoted
        trials so the worker can hand the parent's checkpoint to the
        training function (continuation — the reference restarts promoted
        configs from scratch)."""
        info_dict = {
            "run_budget": run_budget,
            "sample_type": sample_type,
            "sampling_time": time.strftime("%Y-%m-%dT%H:%M:%S"),
        }
        if model_budget is not None:
            info_dict["model_budget"] = model_budget
        if parent_trial_id is not None:
            info_dict["parent_trial_id"] = parent_trial_id
        if run_budget > 0:
            hparams = dict(hparams)
            hparams["budget"] = run_budget
        return Trial(hparams, trial_type="optimization", info_dict=info_dict)

    # min-convention metric access: metrics are negated when the experiment
    # maximizes, so every optimizer can minimize internally
    def get_metrics_dict(self, trial_ids=None):
        sign = -1.0 if self.direction == "max" else 1.0
        out = {}
        for t in self.final_store:
            if trial_ids is None or t.trial_id in trial_ids:
                out[t.trial_id] = (
                    sign * t.final_metric if t.final_metric is not None
                    else None
                )
        return out

    def get_metrics_array(self, include_running=False):
        vals = [m for m in self.get_metrics_dict().values() if m is not None]
        return np.array(vals)

    def ybest(self):
        arr = self.get_metrics_array()
        return float(np.min(arr)) if arr.size else None

    def yworst(self):
        arr = self.get_metrics_array()
        return float(np.max(arr)) if arr.size else None

    def ymean(self):
        arr = self.get_metrics_array()
        return float(np.mean(arr)) if arr.size else None

    def hparams_exist(self, trial):
        """True if a trial with identical hparams (ignoring budget) already
        exists in the trial or final store (parity:
        abstractoptimizer.py:254-295)."""

        def strip(params):
            p = dict(params)
            p.pop("budget", None)
            return Trial._generate_id(p)

        target = strip(trial.params)
        for t in (self.trial_store or {}).values():
            if strip(t.params) == target:
                return True
        for t in self.final_store or []:
            if strip(t.params) == target:
                return True
        return False
