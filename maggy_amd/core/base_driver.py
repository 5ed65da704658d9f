"""Single-run driver: runs the training function inline, no pool.

Parity: /root/reference/maggy/core/experiment_driver/base_driver.py:35-258 +
python_driver.py:104-143 — the "laptop mode" of distribution transparency:
the same train_fn runs in the driver process with a local Reporter, and the
result dict accumulates the returned metric names.
"""
import json
import time

from maggy_amd import util
from maggy_amd.core.environment import Environment
from maggy_amd.core.reporter import Reporter
from maggy_amd.exceptions import EarlyStopException
from maggy_amd.utils.jsonutil import json_default_numpy


class BaseDriver:
    def __init__(self, config, app_id=None, run_id=None):
        self.config = config
        self.name = config.name
        self.description = config.description
        env = Environment.get_instance()
        self.app_id = app_id or env.get_app_id()
        self.run_id = run_id or env.next_run_id(self.app_id)
        self.log_dir = env.get_logdir(self.app_id, self.run_id)
        self.result = {}
        self.job_start = None
        self.job_end = None

    def run_experiment(self, train_fn):
        self.job_start = time.time()
        reporter = Reporter(
            ring=None,
            log_file=self.log_dir + "/executor_0.log",
            worker_id=0,
        )
        trial_dir = self.log_dir + "/single_run"
        Environment.get_instance().mkdir(trial_dir)
        reporter.init_logger(trial_dir + "/output.log")
        hparams = getattr(self.config, "hparams", {}) or {}
        with open(trial_dir + "/.hparams.json", "w") as f:
            f.write(json.dumps(hparams, default=json_default_numpy))
        kwargs = util.build_train_kwargs(
            train_fn,
            model=getattr(self.config, "model", None),
            dataset=getattr(self.config, "dataset", None),
            hparams=hparams,
            reporter=reporter,
        )
        try:
            retval = train_fn(**kwargs)
        except EarlyStopException as e:
            retval = e.metric
        finally:
            reporter.close_logger()
        self.job_end = time.time()

        # accumulate a result dict keyed by returned metric names
        # (parity base_driver.py:221-242)
        if isinstance(retval, dict):
            self.result = dict(retval)
            opt_key = next(iter(retval.keys()))
        elif retval is not None:
            self.result = {"Metric": retval}
            opt_key = "Metric"
        else:
            self.result = {}
            opt_key = None
        if retval is not None:
            util.handle_return_val(
                retval, trial_dir, opt_key, trial_dir + "/output.log")
        duration = util.seconds_to_milliseconds(self.job_end - self.job_start)
        Environment.get_instance().dump(
            json.dumps({"result": self.result, "duration": duration},
                       default=json_default_numpy),
            self.log_dir + "/result.json",
        )
        return self.result
