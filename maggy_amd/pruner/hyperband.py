"""Hyperband pruner: repeated SuccessiveHalving brackets over a worker pool.

Parity: /root/reference/maggy/pruner/hyperband.py:29-594 — geometric
budgets eta^k in [min_budget, max_budget]; n_iterations SH brackets with
per-bracket rung counts n0*eta^-i where n0 = floor(s_max/(rungs+1))*eta^rungs;
``pruning_routine`` returns {"trial_id": parent|None, "budget": b}, "IDLE"
when every active bracket is waiting on running trials, None when all
brackets finished; promotions take the best 1/eta of a completed rung by
the (min-convention) metric getter; original/actual trial-id bookkeeping
preserved so promoted configs rerun at the next budget.  Workers are
pooled: whenever one frees up, waiting brackets with smaller budgets run
first; a new bracket starts only when no active bracket can schedule.
"""
import math

from maggy_amd.pruner.abstract import AbstractPruner


class _Bracket:
    """One SuccessiveHalving bracket (reference SHIteration)."""

    INIT, RUNNING, FINISHED = "INIT", "RUNNING", "FINISHED"

    def __init__(self, bracket_id, n_configs, budgets, metric_getter, log):
        self.bracket_id = bracket_id
        self.state = _Bracket.INIT
        self.n_configs = n_configs        # e.g. [9, 3, 1]
        self.budgets = budgets            # e.g. [1, 3, 9]
        self.n_rungs = len(n_configs)
        self.rung = 0
        self.started = [0] * self.n_rungs  # slots handed to the optimizer
        # rung -> list of {"original": id, "actual": id|None[, "failed",
        #                  "errors"]}
        self.slots = {r: [] for r in range(self.n_rungs)}
        self.metric_getter = metric_getter
        self._log = log
        # bounded retry budget so a deterministically-failing train_fn
        # cannot loop the bracket forever
        self.error_retries_left = sum(n_configs)

    def next_run(self):
        """A schedulable run in this bracket, or None (busy/finished)."""
        quota = self.n_configs[self.rung]
        if self.started[self.rung] < quota:
            if self.rung == 0:
                self.started[0] += 1
                return {"trial_id": None, "budget": self.budgets[0]}
            for slot in self.slots[self.rung]:
                if slot["actual"] is None:
                    self.started[self.rung] += 1
                    return {"trial_id": slot["original"],
                            "budget": self.budgets[self.rung]}
            return None  # promoted slots all handed out, waiting on report
        if self._rung_complete() and self.rung < self.n_rungs - 1:
            self._promote()
            return self.next_run()
        if self._rung_complete() and self.rung == self.n_rungs - 1:
            self.state = _Bracket.FINISHED
            self._log("bracket {} finished".format(self.bracket_id))
        return None

    def _rung_complete(self):
        """Every slot of the current rung created AND finished (a slot
        permanently marked failed counts as finished with worst-case)."""
        if len(self.slots[self.rung]) < self.n_configs[self.rung]:
            return False
        for slot in self.slots[self.rung]:
            if slot.get("failed"):
                continue
            if slot["actual"] is None:
                return False
            if not self.metric_getter([slot["actual"]]):
                return False
        return True

    def _promote(self):
        ids = [s["actual"] for s in self.slots[self.rung]
               if not s.get("failed")]
        metrics = {k: v for k, v in self.metric_getter(ids).items()
                   if v is not None}
        ranked = sorted(metrics, key=metrics.get)
        keep = ranked[: self.n_configs[self.rung + 1]]
        self._log("bracket {} rung {} -> promote {}".format(
            self.bracket_id, self.rung, keep))
        self.rung += 1
        # failed slots shrink the promotion pool; shrink the rung quota so
        # _rung_complete can still be reached
        if len(keep) < self.n_configs[self.rung]:
            self.n_configs[self.rung] = len(keep)
        if not keep and self.rung >= 1:
            self.state = _Bracket.FINISHED
            self._log("bracket {} finished (no promotable trials)".format(
                self.bracket_id))
            return
        for tid in keep:
            self.slots[self.rung].append({"original": tid, "actual": None})

    def on_trial_error(self, trial_id):
        """An assigned trial errored (user exception or worker give-up):
        free its slot so the rung can complete instead of stalling the
        bracket forever.  While the bracket's retry budget lasts, the slot
        is reset to re-run (rung 0 re-samples a fresh config; higher rungs
        re-run the promoted parent); once exhausted, the slot is marked
        failed = finished-with-worst-case and excluded from promotion."""
        for rung in range(self.n_rungs - 1, -1, -1):
            for slot in self.slots[rung]:
                if slot["actual"] == trial_id and not slot.get("failed"):
                    if self.error_retries_left > 0:
                        self.error_retries_left -= 1
                        if rung == 0:
                            self.slots[0].remove(slot)
                        else:
                            slot["actual"] = None
                        self.started[rung] -= 1
                        self._log(
                            "bracket {} rung {}: trial {} errored; slot "
                            "reset for re-run ({} retries left)".format(
                                self.bracket_id, rung, trial_id,
                                self.error_retries_left))
                    else:
                        slot["failed"] = True
                        self._log(
                            "bracket {} rung {}: trial {} errored; retry "
                            "budget exhausted, slot marked failed".format(
                                self.bracket_id, rung, trial_id))
                    return True
        return False

    def report(self, original_trial_id, new_trial_id):
        if self.rung == 0:
            self.slots[0].append(
                {"original": new_trial_id, "actual": new_trial_id})
        else:
            for slot in self.slots[self.rung]:
                if slot["original"] == original_trial_id and \
                        slot["actual"] is None:
                    slot["actual"] = new_trial_id
                    break


class Hyperband(AbstractPruner):
    def __init__(self, min_budget, max_budget, eta=3, n_iterations=None,
                 trial_metric_getter=None):
        super().__init__(trial_metric_getter)
        if min_budget <= 0:
            raise ValueError("min_budget must be > 0")
        if min_budget >= max_budget:
            raise ValueError("max_budget must be > min_budget")
        if eta < 2:
            raise ValueError("eta must be >= 2")
        self.min_budget = min_budget
        self.max_budget = max_budget
        self.eta = eta
        # s_max+1 distinct rung counts, like the reference
        self.max_rungs = int(
            -math.log(min_budget / max_budget) / math.log(eta)) + 1
        if n_iterations is None:
            n_iterations = self.max_rungs
        self.n_pending = n_iterations
        self.budget_ladder = [
            int(max_budget * eta ** -(self.max_rungs - 1 - i))
            for i in range(self.max_rungs)
        ]
        self.brackets = []
        self._updating = None
        for b in range(n_iterations):
            rungs = self.max_rungs - 1 - (b % self.max_rungs)
            n0 = int(math.floor(self.max_rungs / (rungs + 1))
                     * eta ** rungs)
            ns = [max(int(n0 * eta ** -i), 1) for i in range(rungs + 1)]
            budgets = self.budget_ladder[-(rungs + 1):]
            self.brackets.append(
                _Bracket(b, ns, budgets, self._metrics, self._log))

    def _metrics(self, trial_ids):
        return self.trial_metric_getter(trial_ids)

    def initialize(self):
        self._start_next_bracket()

    def _start_next_bracket(self):
        for br in self.brackets:
            if br.state == _Bracket.INIT:
                br.state = _Bracket.RUNNING
                self.n_pending -= 1
                self._log("bracket {} started: n={}, budgets={}".format(
                    br.bracket_id, br.n_configs, br.budgets))
                return True
        return False

    def pruning_routine(self):
        for br in self.brackets:
            if br.state != _Bracket.RUNNING:
                continue
            run = br.next_run()
            if run is not None:
                self._updating = br.bracket_id
                run["bracket"] = br.bracket_id
                return run
        if self.n_pending > 0:
            self._start_next_bracket()
            return self.pruning_routine()
        if self.finished():
            return None
        return "IDLE"

    def report_trial(self, original_trial_id, new_trial_id):
        if self._updating is not None:
            self.brackets[self._updating].report(
                original_trial_id, new_trial_id)
            self._updating = None

    def on_trial_error(self, trial_id):
        """Free the errored trial's bracket slot (see _Bracket.on_trial_error)
        so pruning_routine can hand out a replacement instead of returning
        IDLE forever."""
        for br in self.brackets:
            if br.state == _Bracket.FINISHED:
                continue
            if br.on_trial_error(trial_id):
                return True
        return False

    def finished(self):
        return all(br.state == _Bracket.FINISHED for br in self.brackets)

    def num_trials(self):
        return sum(sum(br.n_configs) for br in self.brackets)

    def on_resume(self, finalized):
        """Rebuild bracket state from persisted trials (info_dict carries
        hb_bracket; budgets map to rungs; parent_trial_id links promoted
        slots).  Rungs are replayed bottom-up: a completed rung re-runs
        the deterministic promotion to regenerate the next rung's slot
        originals, then restored trials claim their slots."""
        by_bracket = {}
        for t in finalized:
            b = (t.info_dict or {}).get("hb_bracket")
            if b is None:
                continue
            by_bracket.setdefault(int(b), []).append(t)
        for bid, trials in sorted(by_bracket.items()):
            br = self.brackets[bid]
            if br.state == _Bracket.INIT:
                br.state = _Bracket.RUNNING
                self.n_pending -= 1
            by_rung = {}
            for t in trials:
                budget = int(t.params.get("budget", br.budgets[0]))
                by_rung.setdefault(br.budgets.index(budget), []).append(t)
            for rung in range(br.n_rungs):
                restored = by_rung.get(rung, [])
                if rung == 0:
                    for t in restored:
                        br.slots[0].append(
                            {"original": t.trial_id, "actual": t.trial_id})
                    br.started[0] = len(br.slots[0])
                else:
                    if not restored and not br._rung_complete():
                        break
                    if br._rung_complete() and br.rung == rung - 1:
                        br._promote()  # regenerates this rung's originals
                    parents = {(t.info_dict or {}).get("parent_trial_id"):
                               t for t in restored}
                    for slot in br.slots[rung]:
                        t = parents.get(slot["original"])
                        if t is not None:
                            slot["actual"] = t.trial_id
                    br.started[rung] = sum(
                        1 for s in br.slots[rung] if s["actual"])
                if not by_rung.get(rung):
                    break
            if br.rung == br.n_rungs - 1 and br._rung_complete():
                br.state = _Bracket.FINISHED
        # make sure at least one bracket is active when work remains
        if not self.finished() and not any(
                br.state == _Bracket.RUNNING for br in self.brackets):
            self._start_next_bracket()
