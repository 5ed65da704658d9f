"""Median stopping rule.

Parity: /root/reference/maggy/earlystop/medianrule.py:27-60 — stop a running
trial if its best metric so far is worse than the median over finalized
trials' running averages truncated at the same step.
"""
import statistics

from maggy_amd.earlystop.abstract import AbstractEarlyStop


class MedianStoppingRule(AbstractEarlyStop):
    @staticmethod
    def earlystop_check(to_check, finalized_trials, direction):
        step = len(to_check.metric_history)
        if step == 0:
            return None
        results = []
        for fin in finalized_trials:
            if len(fin.metric_history) >= step:
                results.append(sum(fin.metric_history[:step]) / float(step))
        if not results:
            return None
        median = statistics.median(results)
        if direction == "max":
            if max(to_check.metric_history) < median:
                return to_check.trial_id
        elif direction == "min":
            if min(to_check.metric_history) > median:
                return to_check.trial_id
        return None
