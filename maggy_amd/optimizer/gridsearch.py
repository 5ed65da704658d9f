"""Grid search over DISCRETE/CATEGORICAL parameters.

Parity: /root/reference/maggy/optimizer/gridsearch.py:33-92 — cartesian
product of the value lists; continuous (DOUBLE/INTEGER) parameters and
pruners are rejected; ``num_trials`` is derived from the grid size.
"""
import itertools

from maggy_amd.optimizer.abstract import AbstractOptimizer


class GridSearch(AbstractOptimizer):
    def __init__(self):
        super().__init__()
        self.config_buffer = []

    def initialize(self):
        if self.pruner is not None:
            raise NotImplementedError("GridSearch does not support pruners")
        for name, ptype in self.searchspace.names().items():
            if ptype not in ("DISCRETE", "CATEGORICAL"):
                raise NotImplementedError(
                    "GridSearch only supports DISCRETE and CATEGORICAL "
                    "parameters; {} is {}".format(name, ptype))
        names = self.searchspace.keys()
        value_lists = [self.searchspace.get(n) for n in names]
        self.config_buffer = [
            dict(zip(names, combo))
            for combo in itertools.product(*value_lists)
        ]
        self.num_trials = len(self.config_buffer)

    def grid_size(self):
        return len(self.config_buffer)

    def get_suggestion(self, trial=None):
        if not self.config_buffer:
            return None
        params = self.config_buffer.pop(0)
        return self.create_trial(params, sample_type="grid")

    def finalize_experiment(self, trials):
        return
