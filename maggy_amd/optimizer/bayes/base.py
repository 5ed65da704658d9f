"""Asynchronous Bayesian optimization base.

Parity: /root/reference/maggy/optimizer/bayes/base.py:88-641 — warmup
buffer of ``num_warmup_trials`` random configs, ``random_fraction``
exploration, per-budget surrogate models, busy-location handling so the
asynchronous workers don't collapse onto one suggestion (constant-liar
imputation or Thompson sampling in the subclasses), duplicate-resample
loop (<=3 attempts).  Internally minimizes: the min-convention metric
getters of AbstractOptimizer negate when direction=="max".
"""
import random

import numpy as np

from maggy_amd.optimizer.abstract import AbstractOptimizer
from maggy_amd.trial import Trial


class BaseAsyncBO(AbstractOptimizer):
    def __init__(self, num_warmup_trials=15, random_fraction=0.33,
                 interim_results=False, interim_results_interval=3,
                 pruner=None, pruner_kwargs=None):
        super().__init__(pruner=pruner, pruner_kwargs=pruner_kwargs)
        self.num_warmup_trials = num_warmup_trials
        self.random_fraction = random_fraction
        self.warmup_buffer = []
        self.models = {}          # budget -> surrogate
        self.sampling_time = []
        self.imputed_metric = "cl_min"  # constant liar default
        # interim-results augmentation (parity: reference base.py z=[x, n],
        # :459-641): train the surrogate on heartbeat metrics at
        # intermediate progress, each observation as [x..., progress]
        self.interim_results = interim_results
        self.interim_results_interval = interim_results_interval

    # -- subclass contract ---------------------------------------------
    def init_model(self, budget=0):
        raise NotImplementedError

    def update_model(self, budget=0):
        raise NotImplementedError

    def sampling_routine(self, budget=0):
        """Return an hparams dict proposed by the surrogate."""
        raise NotImplementedError

    # -- lifecycle -------------------------------------------------------
    def initialize(self):
        for name, ptype in self.searchspace.names().items():
            if ptype not in ("DOUBLE", "INTEGER", "CATEGORICAL", "DISCRETE"):
                raise NotImplementedError(
                    "BO does not support parameter type {}".format(ptype))
        n_warmup = min(self.num_warmup_trials, self.num_trials)
        self.warmup_buffer = self.searchspace.get_random_parameter_values(
            n_warmup)

    def _produced(self):
        return len(self.final_store) + len(self.trial_store)

    def get_suggestion(self, trial=None):
        if trial is not None:
            try:
                self.update_model(budget=self._budget(trial))
            except Exception as e:
                self._log("update_model failed: {}".format(e))
        if self.pruner is not None:
            return self._pruner_suggestion()
        if self._produced() >= self.num_trials:
            return None
        if self.warmup_buffer:
            params = self.warmup_buffer.pop()
            return self.create_trial(params, sample_type="warmup")
        return self._model_or_random(budget=0)

    def _model_or_random(self, budget=0, run_budget=0):
        if random.random() < self.random_fraction or \
                budget not in self.models:
            params = self.searchspace.get_random_parameter_values(1)[0]
            return self.create_trial(params, sample_type="random",
                                     run_budget=run_budget)
        for _ in range(3):  # duplicate-resample loop
            try:
                params = self.sampling_routine(budget=budget)
            except Exception as e:
                self._log("sampling_routine failed: {}".format(e))
                params = None
            if params is None:
                break
            t = Trial(params)
            if not self.hparams_exist(t):
                return self.create_trial(params, sample_type="model",
                                         run_budget=run_budget)
        params = self.searchspace.get_random_parameter_values(1)[0]
        return self.create_trial(params, sample_type="random",
                                 run_budget=run_budget)

    def _pruner_suggestion(self):
        """Multi-fidelity routing via the pruner (parity base.py:187-227)."""
        decision = self.pruner.pruning_routine()
        if decision == "IDLE" or decision is None:
            return decision
        parent_id, budget = decision["trial_id"], decision["budget"]
        if parent_id is None:
            if self.warmup_buffer:
                params = self.warmup_buffer.pop()
                new_trial = self.create_trial(
                    params, sample_type="warmup", run_budget=budget)
            else:
                # BOHB-style: sample from the largest budget with a model
                model_budget = max(
                    (b for b in self.models if b <= budget), default=0)
                new_trial = self._model_or_random(
                    budget=model_budget, run_budget=budget)
        else:
            parent = self._find_trial(parent_id)
            params = dict(parent.params)
            params.pop("budget", None)
            new_trial = self.create_trial(
                params, sample_type="promoted", run_budget=budget,
                parent_trial_id=parent_id)
        if "bracket" in decision:
            new_trial.info_dict["hb_bracket"] = decision["bracket"]
        self.pruner.report_trial(
            original_trial_id=parent_id, new_trial_id=new_trial.trial_id)
        return new_trial

    def _find_trial(self, trial_id):
        for t in self.final_store:
            if t.trial_id == trial_id:
                return t
        if trial_id in (self.trial_store or {}):
            return self.trial_store[trial_id]
        raise KeyError("Trial {} not found".format(trial_id))

    @staticmethod
    def _budget(trial):
        return int(trial.params.get("budget", 0)) if trial else 0

    def on_resume(self, finalized):
        """Experiment resume: the surrogate trains on the preloaded
        final_store; shrink the warmup buffer accordingly and rebuild the
        model.  With a pruner the bracket state is rebuilt too."""
        if self.pruner is not None:
            self.pruner.on_resume(finalized)
        del self.warmup_buffer[:len(finalized)]
        try:
            self.update_model(budget=0)
        except Exception as e:
            self._log("resume update_model failed: {}".format(e))

    def finalize_experiment(self, trials):
        return

    # -- training data ---------------------------------------------------
    def get_XY(self, budget=0, include_busy=True):
        """Build the surrogate training matrix from finalized trials in the
        unit hypercube, min-convention y; busy (running) locations are
        appended with a constant-liar imputed metric (parity
        base.py:400-457).  With a pruner (multi-fidelity), ``budget``
        selects that fidelity's observations (BOHB keeps one model per
        budget); budget=0 or no matches uses every finalized trial."""
        X, y = [], []
        sign = -1.0 if self.direction == "max" else 1.0
        pool = self.final_store
        if budget:
            at_budget = [t for t in self.final_store
                         if t.params.get("budget") == budget]
            if at_budget:
                pool = at_budget
        for t in pool:
            if t.final_metric is None:
                continue
            params = {k: v for k, v in t.params.items() if k != "budget"}
            x = self.searchspace.transform(
                self.searchspace.dict_to_list(params),
                normalize_categorical=True)
            X.append(x)
            y.append(sign * t.final_metric)
        n_fin = len(y)
        if self.interim_results:
            # augment each observation with a progress coordinate: final
            # metrics at 1.0, heartbeat metrics at their fractional step
            for i in range(n_fin):
                X[i] = X[i] + [1.0]
            for t in pool:
                if t.final_metric is None or not t.metric_history:
                    continue
                hist = t.metric_history
                T = len(hist)
                params = {k: v for k, v in t.params.items()
                          if k != "budget"}
                xb = self.searchspace.transform(
                    self.searchspace.dict_to_list(params),
                    normalize_categorical=True)
                for s in range(0, T - 1, self.interim_results_interval):
                    X.append(xb + [(s + 1) / T])
                    y.append(sign * hist[s])
        if include_busy and n_fin:
            # kriging believer imputes each busy location with the current
            # model's posterior mean (reference gp.py:329-373); constant
            # liar uses a fixed pessimistic/optimistic value.  KB falls
            # back to cl_min before the first model fit.
            kb_model = None
            if self.imputed_metric == "kb":
                kb_model = self.models.get(budget)
                if kb_model is not None and not hasattr(
                        kb_model, "X_train_"):
                    kb_model = None
            liar = {
                "cl_min": min(y[:n_fin]), "cl_max": max(y[:n_fin]),
                "cl_mean": sum(y[:n_fin]) / n_fin, "kb": min(y[:n_fin]),
            }[self.imputed_metric]
            for t in self.trial_store.values():
                params = {k: v for k, v in t.params.items()
                          if k != "budget"}
                try:
                    x = self.searchspace.transform(
                        self.searchspace.dict_to_list(params),
                        normalize_categorical=True)
                except (KeyError, ValueError):
                    continue
                if self.interim_results:
                    x = x + [1.0]
                X.append(x)
                if kb_model is not None:
                    y.append(float(kb_model.predict(
                        np.asarray([x], dtype=float))[0]))
                else:
                    y.append(liar)
        return np.asarray(X, dtype=float), np.asarray(y, dtype=float), n_fin
