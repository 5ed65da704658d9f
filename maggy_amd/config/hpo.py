"""Hyperparameter-optimization config.

Parity: /root/reference/maggy/config/hyperparameter_optimization.py:33-93 —
same fields and defaults (optimization_key="Metric", direction="max",
es_interval=1, es_min=10, es_policy="median"). The reference requires Spark;
here the experiment runs on the in-process trial pool, one trial per GPU
(``num_workers`` defaults to the number of visible GPUs, or 1 on CPU).
"""
from maggy_amd.config.lagom import LagomConfig


class HyperparameterOptConfig(LagomConfig):
    def __init__(
        self,
        num_trials,
        optimizer,
        searchspace,
        optimization_key="Metric",
        direction="max",
        es_interval=1,
        es_min=10,
        es_policy="median",
        name="HPOptimization",
        description="",
        hb_interval=1,
        model=None,
        dataset=None,
        num_workers=None,
    ):
        super().__init__(name=name, description=description, hb_interval=hb_interval)
        if not num_trials or num_trials < 1:
            raise ValueError("num_trials must be >= 1, got {}".format(num_trials))
        self.num_trials = num_trials
        self.optimizer = optimizer
        self.optimization_key = optimization_key
        self.searchspace = searchspace
        self.direction = direction
        self.es_policy = es_policy
        self.es_interval = es_interval
        self.es_min = es_min
        self.model = model
        self.dataset = dataset
        # number of concurrent trial workers (None -> one per visible GPU)
        self.num_workers = num_workers
