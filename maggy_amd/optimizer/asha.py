"""ASHA — Asynchronous Successive Halving (arXiv 1810.05934).

Parity: /root/reference/maggy/optimizer/asha.py:29-169 — rung dict, top-1/eta
promotion from the highest possible rung, random base-rung configs at
``resource_min``, stop when the max rung is populated.  One behavioral fix
over the reference: ``_top_k`` sorts by the experiment direction instead of
always descending (the reference silently assumed direction=="max").
"""
import math

from maggy_amd.optimizer.abstract import AbstractOptimizer
from maggy_amd.trial import Trial


class Asha(AbstractOptimizer):
    def __init__(self, reduction_factor=2, resource_min=1, resource_max=4):
        super().__init__()
        if not isinstance(reduction_factor, int) or reduction_factor < 2:
            raise ValueError(
                "reduction_factor has to be an integer >= 2: {}".format(
                    reduction_factor))
        if not isinstance(resource_min, int) or not isinstance(
                resource_max, int):
            raise ValueError("resource_min/resource_max must be int")
        if resource_min >= resource_max:
            raise ValueError(
                "resource_min ({}) must be < resource_max ({})".format(
                    resource_min, resource_max))
        self.reduction_factor = reduction_factor
        self.resource_min = resource_min
        self.resource_max = resource_max
        self.rungs = {}
        self.promoted = {}
        self.max_rung = 0

    def initialize(self):
        self.rungs = {0: []}
        self.promoted = {0: []}
        self.max_rung = int(math.floor(math.log(
            self.resource_max / self.resource_min, self.reduction_factor)))
        if self.num_trials < self.reduction_factor ** (self.max_rung + 1):
            raise ValueError(
                "num_trials ({}) must be >= reduction_factor ** (max_rung+1) "
                "= {}".format(
                    self.num_trials,
                    self.reduction_factor ** (self.max_rung + 1)))

    def get_suggestion(self, trial=None):
        # stopping criterion: a trial reached the max rung (checked for
        # every request so a resumed-complete experiment terminates
        # without sampling fresh base-rung trials)
        if self.rungs.get(self.max_rung):
            return None
        if trial is not None:
            # try to promote, scanning rungs from high to low
            for k in range(self.max_rung - 1, -1, -1):
                if k not in self.rungs:
                    continue
                rung_finished = len([
                    t for t in self.rungs[k] if t.status == Trial.FINALIZED])
                quota = rung_finished // self.reduction_factor
                if quota - len(self.promoted.get(k, [])) <= 0:
                    continue
                candidates = self._top_k(k, quota)
                promotable = [
                    t for t in candidates
                    if t.trial_id not in self.promoted.get(k, [])]
                if not promotable:
                    continue
                new_rung = k + 1
                old_trial = promotable[0]
                params = dict(old_trial.params)
                budget = self.resource_min * (
                    self.reduction_factor ** new_rung)
                params.pop("budget", None)
                promote_trial = self.create_trial(
                    params, sample_type="promoted", run_budget=budget,
                    parent_trial_id=old_trial.trial_id)
                self.rungs.setdefault(new_rung, []).append(promote_trial)
                self.promoted.setdefault(k, []).append(old_trial.trial_id)
                self._log("promote {} -> rung {} (budget {})".format(
                    old_trial.trial_id, new_rung, budget))
                return promote_trial
        # fall through: random config in the base rung
        params = self.searchspace.get_random_parameter_values(1)[0]
        new_trial = self.create_trial(
            params, sample_type="random", run_budget=self.resource_min)
        self.rungs[0].append(new_trial)
        return new_trial

    def _top_k(self, rung_k, number):
        if number <= 0:
            return []
        finished = [
            t for t in self.rungs[rung_k]
            if t.status == Trial.FINALIZED and t.final_metric is not None]
        finished.sort(
            key=lambda t: t.final_metric, reverse=(self.direction == "max"))
        return finished[:number]

    def on_resume(self, finalized):
        """Rebuild the rung state from persisted trials: the rung index is
        log_eta(budget/resource_min) (stamped into hparams by
        create_trial), and a trial whose info_dict carries
        parent_trial_id was created by promoting that parent."""
        for t in finalized:
            budget = int(t.params.get("budget", self.resource_min))
            rung = int(round(math.log(
                budget / self.resource_min, self.reduction_factor)))
            self.rungs.setdefault(rung, []).append(t)
            parent = (t.info_dict or {}).get("parent_trial_id")
            if parent is not None and rung > 0:
                self.promoted.setdefault(rung - 1, []).append(parent)

    def finalize_experiment(self, trials):
        return
