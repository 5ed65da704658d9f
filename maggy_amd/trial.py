"""Trial: the unit of work handed to a GPU worker.

Parity: /root/reference/maggy/trial.py:24-176 — same status machine
(PENDING/SCHEDULED/RUNNING/ERROR/FINALIZED), the same content-addressed
16-char md5 trial id over the sorted-params JSON (so the reference's test
vector "3d1cc9fdb1d4d001" for {"param1": 5, "param2": "ada"} still holds),
and the same JSON (de)serialization schema so trial.json artifacts are
drop-in compatible.
"""
import hashlib
import json
import threading
import time

from maggy_amd.utils.jsonutil import json_default_numpy


class Trial:
    PENDING = "PENDING"
    SCHEDULED = "SCHEDULED"
    RUNNING = "RUNNING"
    ERROR = "ERROR"
    FINALIZED = "FINALIZED"

    def __init__(self, params, trial_type="optimization", info_dict=None):
        self.params = params
        self.trial_type = trial_type
        self.trial_id = Trial._generate_id(params)
        self.status = Trial.PENDING
        self.early_stop = False
        self.final_metric = None
        # metric_history: deduped per-step values in arrival order
        self.metric_history = []
        self.step_history = []
        self.metric_dict = {}
        self.duration = None
        self.start = None
        self.lock = threading.RLock()
        self.info_dict = info_dict if info_dict is not None else {}

    def get_early_stop(self):
        with self.lock:
            return self.early_stop

    def set_early_stop(self):
        with self.lock:
            self.early_stop = True

    def append_metric(self, metric_data):
        """Append a heartbeat metric record ``{"value": v, "step": s}``.

        Returns the step if it was a new unique step, else None (parity with
        reference trial.py:93-108: dedup by step).
        """
        with self.lock:
            if (
                metric_data.get("step") is not None
                and metric_data["step"] not in self.metric_dict
                and metric_data.get("value") is not None
            ):
                self.metric_dict[metric_data["step"]] = metric_data["value"]
                self.metric_history.append(metric_data["value"])
                self.step_history.append(metric_data["step"])
                return metric_data["step"]
            return None

    @classmethod
    def _generate_id(cls, params):
        """16-char truncated md5 of the sorted-params JSON (stable across
        processes; same function as reference trial.py:111-136)."""
        if not isinstance(params, dict):
            raise ValueError("Hyperparameters need to be a dictionary.")
        if not all(isinstance(k, str) for k in params.keys()):
            raise ValueError("All hyperparameter names have to be strings.")
        return hashlib.md5(
            json.dumps(params, sort_keys=True).encode("utf-8")
        ).hexdigest()[:16]

    def to_dict(self):
        obj_dict = {"__class__": self.__class__.__name__}
        temp = self.__dict__.copy()
        temp.pop("lock")
        temp.pop("start")
        obj_dict.update(temp)
        return obj_dict

    def to_json(self):
        return json.dumps(self.to_dict(), default=json_default_numpy)

    @classmethod
    def from_json(cls, json_str):
        d = json.loads(json_str)
        if d.get("__class__", None) != "Trial":
            raise ValueError("json_str is not a Trial object.")
        instance = None
        if d.get("params", None) is not None:
            instance = cls(d["params"])
            instance.trial_id = d["trial_id"]
            instance.status = d["status"]
            instance.early_stop = d.get("early_stop", False)
            instance.final_metric = d["final_metric"]
            instance.metric_history = d["metric_history"]
            instance.step_history = d.get("step_history", [])
            instance.metric_dict = {
                int(k): v for k, v in d.get("metric_dict", {}).items()
            }
            instance.duration = d["duration"]
            instance.info_dict = d.get("info_dict", {})
        return instance

    def mark_running(self):
        with self.lock:
            self.status = Trial.RUNNING
            if self.start is None:
                self.start = time.time()

    def __repr__(self):
        return "Trial({}, status={}, final_metric={})".format(
            self.trial_id, self.status, self.final_metric
        )
