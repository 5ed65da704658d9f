"""Random search.

Parity: /root/reference/maggy/optimizer/randomsearch.py:24-113 — pre-samples
``num_trials`` configurations into a buffer; with a pruner attached, routes
through ``pruner.pruning_routine()`` to run promoted or random trials at the
pruner's budgets.
"""
from maggy_amd.optimizer.abstract import AbstractOptimizer


class RandomSearch(AbstractOptimizer):
    def __init__(self, pruner=None, pruner_kwargs=None):
        super().__init__(pruner=pruner, pruner_kwargs=pruner_kwargs)
        self.config_buffer = []

    def initialize(self):
        for name, ptype in self.searchspace.names().items():
            if ptype not in ("DOUBLE", "INTEGER", "DISCRETE", "CATEGORICAL"):
                raise NotImplementedError(
                    "RandomSearch does not support parameter type {}".format(
                        ptype))
        if self.pruner is None:
            self.config_buffer = self.searchspace.get_random_parameter_values(
                self.num_trials)

    def get_suggestion(self, trial=None):
        if self.pruner is not None:
            return self._pruner_suggestion(trial)
        if len(self.config_buffer) == 0:
            return None
        params = self.config_buffer.pop()
        return self.create_trial(params, sample_type="random")

    def _pruner_suggestion(self, trial):
        """Promoted-vs-random routine (parity randomsearch.py:47-90)."""
        decision = self.pruner.pruning_routine()
        if decision == "IDLE":
            return "IDLE"
        if decision is None:
            return None
        parent_id, budget = decision["trial_id"], decision["budget"]
        if parent_id is None:
            params = self.searchspace.get_random_parameter_values(1)[0]
            new_trial = self.create_trial(
                params, sample_type="random", run_budget=budget)
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

    def on_resume(self, finalized):
        """Experiment resume: skip as many pre-sampled configs as have
        already finalized; with a pruner, the pruner rebuilds its bracket
        state from the persisted records."""
        if self.pruner is not None:
            self.pruner.on_resume(finalized)
            return
        del self.config_buffer[:len(finalized)]

    def finalize_experiment(self, trials):
        return
