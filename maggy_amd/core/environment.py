"""Local-filesystem environment: experiment directories and artifacts.

Parity: /root/reference/maggy/core/environment/base.py:25 (BaseEnv) plus the
app/run-id management of /root/reference/maggy/experiment/experiment_python.py
:71-72 and util.py:216-243. The reference's Hopsworks/Databricks environments
collapse to this single local-FS environment; the ``Environment`` seam is
kept so artifact paths stay swappable.

Directory tree produced per experiment (identical to the reference, §5.4 of
SURVEY.md):

    <base>/<app_id>/<run_id>/
        maggy.log                  driver log
        optimizer.log              controller log
        pruner.log                 (when a pruner is active)
        executor_<id>.log          per-worker system log
        <trial_id>/
            .hparams.json
            output.log
            .outputs.json
            .metric
            trial.json
        result.json
        maggy.json
"""
import json
import os
import threading
import time

from maggy_amd.utils.jsonutil import json_default_numpy

_env_lock = threading.Lock()
_instance = None


class Environment:
    """Local FS environment singleton (parity: EnvSing.get_instance())."""

    def __init__(self, base_dir=None):
        self.base_dir = base_dir or os.environ.get(
            "MAGGY_LOG_DIR", os.path.join(os.getcwd(), "maggy_experiments")
        )
        self._app_id = None

    # -- singleton ------------------------------------------------------
    @staticmethod
    def get_instance():
        global _instance
        with _env_lock:
            if _instance is None:
                _instance = Environment()
            return _instance

    @staticmethod
    def set_instance(env):
        global _instance
        with _env_lock:
            _instance = env

    # -- app / run ids --------------------------------------------------
    def get_app_id(self):
        """Synthesized Spark-style app id (reference experiment_python.py:71)."""
        if self._app_id is None:
            self._app_id = "application_{}_0001".format(int(time.time()))
        return self._app_id

    def next_run_id(self, app_id):
        app_dir = os.path.join(self.base_dir, app_id)
        os.makedirs(app_dir, exist_ok=True)
        existing = [int(d) for d in os.listdir(app_dir) if d.isdigit()]
        return (max(existing) + 1) if existing else 1

    def get_logdir(self, app_id, run_id):
        d = os.path.join(self.base_dir, app_id, str(run_id))
        os.makedirs(d, exist_ok=True)
        return d

    # -- file ops -------------------------------------------------------
    def exists(self, path):
        return os.path.exists(path)

    def mkdir(self, path):
        os.makedirs(path, exist_ok=True)

    def dump(self, data, path):
        os.makedirs(os.path.dirname(path), exist_ok=True)
        mode = "wb" if isinstance(data, bytes) else "w"
        with open(path, mode) as f:
            f.write(data)

    def load(self, path):
        with open(path, "r") as f:
            return f.read()

    def open_file(self, path, flags="w"):
        os.makedirs(os.path.dirname(path), exist_ok=True)
        return open(path, flags)

    def dump_json(self, obj, path):
        self.dump(json.dumps(obj, default=json_default_numpy, indent=2), path)
