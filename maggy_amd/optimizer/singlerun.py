"""SingleRun: ``num_trials`` empty-parameter trials (optimizer=None).

Parity: /root/reference/maggy/optimizer/singlerun.py:21-37.  Each trial gets
a unique synthetic param so trial ids/dirs don't collide.
"""
from maggy_amd.optimizer.abstract import AbstractOptimizer
from maggy_amd.trial import Trial


class SingleRun(AbstractOptimizer):
    def __init__(self):
        super().__init__()
        self._produced = 0

    def initialize(self):
        self._produced = 0

    def get_suggestion(self, trial=None):
        if self._produced >= self.num_trials:
            return None
        self._produced += 1
        return Trial({"run": self._produced}, trial_type="optimization")

    def finalize_experiment(self, trials):
        return
