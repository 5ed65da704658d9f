"""Experiment drivers: the control plane of the trial pool.

Replaces the reference's Spark-driver + RPC-server + digestion-thread stack
(/root/reference/maggy/core/experiment_driver/spark_driver.py:103-287 and
optimization_driver.py:433-592) with a single-threaded event loop over the
worker pipes and shared-memory metric rings.  The scheduling state machine
is preserved exactly:

  REG    -> first assignment (or IDLE requeue / experiment done)
  METRIC -> append to trial history; every ``es_interval`` unique steps
            after ``es_min`` finalized trials, run the early-stop policy and
            flag the trial's stop word
  FINAL  -> finalize trial, dump trial.json, update result bookkeeping,
            hand the worker the controller's next suggestion (or IDLE/GSTOP)
  dead worker -> respawn + re-queue its trial (the reference's BLACK path,
            rpc.py:415-437)
  IDLE   -> controller had no trial ready; retried after 0.1 s
"""
import json
import threading
import time

from maggy_amd import util
from maggy_amd.core import messages as M
from maggy_amd.core.environment import Environment
from maggy_amd.core.pool import TrialPool
from maggy_amd.core.shm import trial_tag
from maggy_amd.earlystop import MedianStoppingRule, NoStoppingRule
from maggy_amd.trial import Trial
from maggy_amd.utils.jsonutil import json_default_numpy

IDLE_RETRY_S = 0.1
PROGRESS_INTERVAL_S = 2.0


class OptimizationDriver:
    """Async HPO driver over the GPU trial pool."""

    def __init__(self, config, app_id=None, run_id=None, num_workers=None,
                 gpu_ids=None):
        from maggy_amd.optimizer import resolve_controller

        self.config = config
        self.name = config.name
        self.description = config.description
        self.num_trials = getattr(config, "num_trials", 1)
        self.optimization_key = getattr(config, "optimization_key", "Metric")
        self.direction = getattr(config, "direction", "max").lower()
        if self.direction not in ("max", "min"):
            raise ValueError("direction must be 'max' or 'min'")
        self.es_interval = getattr(config, "es_interval", 1)
        self.es_min = getattr(config, "es_min", 10)
        self.searchspace = getattr(config, "searchspace", None)

        env = Environment.get_instance()
        self.app_id = app_id or env.get_app_id()
        self.run_id = run_id or env.next_run_id(self.app_id)
        self.log_dir = env.get_logdir(self.app_id, self.run_id)
        self._log_fd = open(self.log_dir + "/maggy.log", "w")

        # worker count: explicit > config > one per GPU > 1
        if num_workers is None:
            num_workers = getattr(config, "num_workers", None)
        if num_workers is None:
            num_workers = max(1, util.num_gpus())
        # cap concurrency at num_trials (reference optimization_driver.py:81)
        self.num_workers = min(num_workers, self.num_trials)
        if gpu_ids is None:
            n_gpu = util.num_gpus()
            gpu_ids = [i % n_gpu if n_gpu else None
                       for i in range(self.num_workers)]
        self.gpu_ids = gpu_ids

        # trial stores (parity: driver _trial_store / _final_store)
        self._trial_store = {}
        self._final_store = []
        self._error_store = []
        self._tag_to_trial = {}

        self.controller = resolve_controller(
            getattr(config, "optimizer", None), self
        )
        es_policy = getattr(config, "es_policy", "median")
        self.earlystop = self._resolve_es(es_policy)
        self.experiment_done = False
        self.result = {}
        self.maggy_log = ""
        self.executor_logs = ""  # jupyter-style log stream (parity LOG msg)
        self._log_stream_lock = threading.Lock()
        # live-progress callback(status_string, new_logs), invoked from
        # the event loop every PROGRESS_INTERVAL_S (parity: the LOG
        # snapshots the reference served to Jupyter mid-run,
        # rpc.py:490-502); set via experiment.lagom(..., progress=...)
        self.progress_cb = None
        self._last_progress = 0.0
        self.job_start = None
        self.job_end = None
        self.duration = None
        self._idle_workers = {}  # worker_id -> idle_start time
        # scheduling telemetry: first trial assignment / last finalization
        self.first_assign_ts = None
        self.last_final_ts = None

    # ------------------------------------------------------------------
    @staticmethod
    def _resolve_es(policy):
        if policy is None or policy == "none":
            return NoStoppingRule
        if policy == "median":
            return MedianStoppingRule
        if isinstance(policy, type):
            return policy
        raise ValueError("Unknown early-stop policy: {}".format(policy))

    def log(self, msg):
        try:
            self._log_fd.write(
                "{}: {}\n".format(time.strftime("%Y-%m-%dT%H:%M:%S"), msg)
            )
            self._log_fd.flush()
        except ValueError:
            pass

    # -- trial store ----------------------------------------------------
    def add_trial(self, trial):
        self._trial_store[trial.trial_id] = trial
        self._tag_to_trial[trial_tag(trial.trial_id)] = trial

    def get_trial(self, trial_id):
        return self._trial_store.get(trial_id)

    # ------------------------------------------------------------------
    def resume_from(self, exp_dir):
        """Preload finalized trials from a previous experiment directory
        (the reference left this latent: Trial.from_json existed but was
        never called).  Call before run_experiment; the
        controller is told via ``on_resume`` so it does not re-run the
        completed configurations."""
        from maggy_amd.utils.checkpoint import load_finished_trials

        loaded = [t for t in load_finished_trials(exp_dir)
                  if t is not None and t.status == Trial.FINALIZED]
        for t in loaded:
            self._final_store.append(t)
            self._update_result(t)
        self._resumed = len(loaded)
        self.log("resumed {} finalized trials from {}".format(
            len(loaded), exp_dir))
        return len(loaded)

    def run_experiment(self, train_fn, payload_extra=None):
        """Run the experiment to completion; returns the result dict."""
        self.job_start = time.time()
        self.controller._initialize(exp_dir=self.log_dir)
        resumed = getattr(self, "_resumed", 0)
        if resumed:
            if hasattr(self.controller, "on_resume"):
                self.controller.on_resume(self._final_store)
            else:
                raise NotImplementedError(
                    "{} does not support resume".format(
                        self.controller.name()))
        payload = {
            "train_fn": train_fn,
            "model": getattr(self.config, "model", None),
            "dataset": getattr(self.config, "dataset", None),
            "optimization_key": self.optimization_key,
            "experiment_type": self.controller.experiment_type,
        }
        if payload_extra:
            payload.update(payload_extra)
        self.pool = TrialPool(
            self.num_workers, self.log_dir, payload, gpu_ids=self.gpu_ids
        )
        self.pool.start()
        try:
            self._event_loop()
        finally:
            self.pool.shutdown()
        self.finalize(time.time())
        return self.result

    # ------------------------------------------------------------------
    def _event_loop(self):
        pool = self.pool
        while True:
            for w, msg in pool.poll_messages():
                kind = msg[0]
                if kind == M.REG:
                    w.registered = True
                    self.log("worker {} registered (pid {})".format(
                        w.worker_id, msg[2]))
                    if not self._assign_pending(w):
                        self._assign_next(w)
                elif kind == M.FINAL:
                    self._handle_final(w, msg)
                elif kind == M.ERROR:
                    self._handle_error(w, msg)
                elif kind == M.LOG:
                    with self._log_stream_lock:
                        self.executor_logs += msg[2]
            self._handle_metrics(pool.drain_metrics())
            self._report_ring_drops()
            self._emit_progress()
            self._retry_idle()
            self._reap_dead()
            if self.experiment_done and self._all_workers_free():
                break
            if not any(w.process is not None and w.process.is_alive()
                       for w in pool.workers):
                # a worker that died AFTER this tick's _reap_dead() is
                # respawned on the next tick — give up only when no
                # worker is respawnable any more (TOCTOU flake otherwise)
                if all(w.process is None or w.respawns >= 3
                       for w in pool.workers):
                    from maggy_amd.exceptions import WorkerCrashError

                    raise WorkerCrashError("all", "all trial workers died")

    def _all_workers_free(self):
        return all(w.trial_id is None for w in self.pool.workers)

    # -- assignment ------------------------------------------------------
    def controller_get_next(self, trial=None):
        return self.controller.get_suggestion(trial)

    def _assign_next(self, w, finished_trial=None):
        """FINAL/REG hand-off (parity optimization_driver.py:485-592)."""
        if self.experiment_done:
            w.trial_id = None
            return
        trial = self.controller_get_next(finished_trial)
        if trial is None:
            w.trial_id = None
            self.experiment_done = True
        elif trial == "IDLE":
            w.trial_id = None
            self._idle_workers[w.worker_id] = time.time()
        else:
            trial.start = time.time()
            if self.first_assign_ts is None:
                self.first_assign_ts = trial.start
            trial.status = Trial.SCHEDULED
            self.add_trial(trial)
            self.pool.assign(w, trial)

    def _retry_idle(self):
        if not self._idle_workers:
            return
        now = time.time()
        for wid in list(self._idle_workers):
            if now - self._idle_workers[wid] < IDLE_RETRY_S:
                continue
            w = self.pool.workers[wid]
            del self._idle_workers[wid]
            if w.trial_id is None and w.registered:
                self._assign_next(w)

    # -- message handlers ------------------------------------------------
    def _handle_final(self, w, msg):
        _, worker_id, trial_id, opt_val, dur, early_stopped, logs = msg
        # the metric ring and the control pipe are independent channels: drain
        # the worker's ring first so the trial's full metric history is
        # recorded before it is finalized
        if w.ring is not None:
            self._handle_metrics(
                [(w, tag, step, value) for tag, step, value in w.ring.drain()]
            )
        trial = self.get_trial(trial_id)
        if trial is None:
            self.log("FINAL for unknown trial {}".format(trial_id))
            self._assign_next(w)
            return
        with trial.lock:
            trial.status = Trial.FINALIZED
            trial.final_metric = opt_val
            trial.early_stop = bool(early_stopped)
            trial.duration = util.seconds_to_milliseconds(dur)
        self._final_store.append(trial)
        self._trial_store.pop(trial_id, None)
        w.trial_id = None
        self.last_final_ts = time.time()
        if logs:
            with self._log_stream_lock:
                self.executor_logs += logs
        self._update_result(trial)
        self.maggy_log = self.log_string()
        self.log(self.maggy_log)
        Environment.get_instance().dump(
            trial.to_json(), "{}/{}/trial.json".format(self.log_dir, trial_id)
        )
        self._assign_next(w, finished_trial=trial)

    def _handle_error(self, w, msg):
        _, worker_id, trial_id, tb = msg
        trial = self.get_trial(trial_id)
        self.log("trial {} errored on worker {}:\n{}".format(
            trial_id, worker_id, tb))
        if trial is not None:
            with trial.lock:
                trial.status = Trial.ERROR
            self._error_store.append(trial)
            self._trial_store.pop(trial_id, None)
            self._notify_controller_error(trial)
        w.trial_id = None
        self._assign_next(w)

    def _notify_controller_error(self, trial):
        """Tell the controller a trial died so budget-based schedulers
        (Hyperband brackets) free the slot instead of waiting forever."""
        try:
            if hasattr(self.controller, "on_trial_error"):
                self.controller.on_trial_error(trial)
        except Exception as e:
            self.log("controller on_trial_error failed: {}".format(e))

    def _handle_metrics(self, records):
        """Metric-stream digestion + early-stop policy (parity
        optimization_driver.py:433-471)."""
        for w, tag, step, value in records:
            trial = self._tag_to_trial.get(tag)
            if trial is None or trial.status == Trial.FINALIZED:
                continue
            if trial.status == Trial.SCHEDULED:
                trial.status = Trial.RUNNING
            new_step = trial.append_metric({"value": value, "step": int(step)})
            if (
                new_step is not None
                and new_step != 0
                and self.earlystop is not NoStoppingRule
                and len(self._final_store) > self.es_min
                and new_step % self.es_interval == 0
                and not trial.get_early_stop()
            ):
                try:
                    to_stop = self.earlystop.earlystop_check(
                        trial, self._final_store, self.direction
                    )
                except Exception as e:
                    self.log("earlystop_check failed: {}".format(e))
                    to_stop = None
                if to_stop is not None:
                    self.log("Trials to stop: {}".format(to_stop))
                    t = self.get_trial(to_stop)
                    if t is not None:
                        t.set_early_stop()
                    self.pool.request_stop(to_stop)

    def _report_ring_drops(self):
        """Surface metric-ring overruns in maggy.log: a fast-broadcasting
        trial that wraps the ring loses its oldest heartbeat records, which
        the median early-stop rule may have needed — make that observable
        instead of silent."""
        reported = getattr(self, "_drops_reported", {})
        for w in self.pool.workers:
            if w.ring is None:
                continue
            prev = reported.get(w.worker_id, 0)
            if w.ring.dropped > prev:
                self.log(
                    "worker {}: metric ring overran; {} records dropped "
                    "({} total) — consider fewer broadcasts or more ring "
                    "slots".format(w.worker_id, w.ring.dropped - prev,
                                   w.ring.dropped))
                reported[w.worker_id] = w.ring.dropped
        self._drops_reported = reported

    def _reap_dead(self):
        """Worker-crash recovery (parity: BLACK re-binding, rpc.py:415-437)."""
        for w in self.pool.reap():
            lost_trial_id = w.trial_id
            self.log("worker {} died (exit {}); respawning".format(
                w.worker_id, w.process.exitcode))
            if w.respawns >= 3:
                self.log("worker {} died {} times; not respawning".format(
                    w.worker_id, w.respawns))
                w.process = None
                # the held trial can never finish: mark it ERROR so
                # budget-based controllers (ASHA rungs) don't wait forever
                if lost_trial_id is not None:
                    trial = self.get_trial(lost_trial_id)
                    if trial is not None:
                        with trial.lock:
                            trial.status = Trial.ERROR
                        self._error_store.append(trial)
                        self._trial_store.pop(lost_trial_id, None)
                        self._notify_controller_error(trial)
                w.trial_id = None
                continue
            self.pool.respawn(w)
            if lost_trial_id is not None:
                trial = self.get_trial(lost_trial_id)
                if trial is not None:
                    # re-assign the same trial to the respawned worker once
                    # it re-registers; reset its metric history
                    with trial.lock:
                        trial.status = Trial.SCHEDULED
                        trial.metric_history = []
                        trial.step_history = []
                        trial.metric_dict = {}
                    w.trial_id = lost_trial_id
                    self._pending_reassign = getattr(
                        self, "_pending_reassign", {})
                    self._pending_reassign[w.worker_id] = lost_trial_id

    def _assign_pending(self, w):
        pending = getattr(self, "_pending_reassign", {})
        trial_id = pending.pop(w.worker_id, None)
        if trial_id is not None:
            trial = self.get_trial(trial_id)
            if trial is not None:
                self.pool.assign(w, trial)
                return True
        return False

    # -- result bookkeeping (parity optimization_driver.py:344-406) ------
    def _update_result(self, trial):
        metric = trial.final_metric
        params = dict(trial.params)
        params.pop("dataset_function", None)
        params.pop("model_function", None)
        tid = trial.trial_id
        num_epochs = len(trial.metric_history)
        if self.result.get("best_id") is None:
            self.result = {
                "best_id": tid, "best_val": metric, "best_config": params,
                "worst_id": tid, "worst_val": metric, "worst_config": params,
                "avg": metric, "metric_list": [metric], "num_trials": 1,
                "early_stopped": 1 if trial.early_stop else 0,
                "num_epochs": num_epochs, "trial_id": tid,
            }
            return
        better = (lambda a, b: a > b) if self.direction == "max" \
            else (lambda a, b: a < b)
        if metric is not None:
            if self.result["best_val"] is None or \
                    better(metric, self.result["best_val"]):
                self.result.update(
                    best_val=metric, best_id=tid, best_config=params)
            if self.result["worst_val"] is None or \
                    better(self.result["worst_val"], metric):
                self.result.update(
                    worst_val=metric, worst_id=tid, worst_config=params)
        self.result["metric_list"].append(metric)
        self.result["num_trials"] += 1
        numeric = [m for m in self.result["metric_list"] if m is not None]
        self.result["avg"] = sum(numeric) / len(numeric) if numeric else None
        if trial.early_stop:
            self.result["early_stopped"] += 1

    def _emit_progress(self):
        if self.progress_cb is None:
            return
        now = time.time()
        if now - self._last_progress < PROGRESS_INTERVAL_S:
            return
        self._last_progress = now
        try:
            self.progress_cb(*self.get_logs())
        except Exception as e:
            self.log("progress callback failed: {}".format(e))

    def get_logs(self):
        """Drain the accumulated executor log stream plus a progress
        snapshot (parity: LOG request, rpc.py:490-502).  Thread-safe:
        callable mid-run from another thread (lagom_async handle)."""
        with self._log_stream_lock:
            logs = self.executor_logs
            self.executor_logs = ""
        return self.log_string(), logs

    def log_string(self):
        return (
            "Maggy Optimization {}/{} ({}) {} - BEST {} - metric {}".format(
                self.result.get("num_trials", 0), self.num_trials,
                self.result.get("early_stopped", 0),
                util.progress_bar(self.result.get("num_trials", 0),
                                  self.num_trials),
                json.dumps(self.result.get("best_config", {}),
                           default=json_default_numpy),
                self.result.get("best_val"),
            )
        )

    # -- finalize (parity optimization_driver.py:235-342) ----------------
    def finalize(self, job_end):
        self.job_end = job_end
        self.duration = util.seconds_to_milliseconds(
            self.job_end - self.job_start)
        try:
            self.controller._finalize_experiment(self._final_store)
        except Exception as e:
            self.log("controller finalize failed: {}".format(e))
        env = Environment.get_instance()
        env.dump(
            json.dumps(self.result, default=json_default_numpy),
            self.log_dir + "/result.json",
        )
        env.dump(self.experiment_json(), self.log_dir + "/maggy.json")
        summary = (
            "\n------ {} Results ------ direction({})\n"
            "BEST combination {} -- metric {}\n"
            "WORST combination {} -- metric {}\n"
            "AVERAGE metric -- {}\n"
            "EARLY STOPPED Trials -- {}\n"
            "Total job time {} ms\n".format(
                self.controller.name(), self.direction,
                json.dumps(self.result.get("best_config", {}),
                           default=json_default_numpy),
                self.result.get("best_val"),
                json.dumps(self.result.get("worst_config", {}),
                           default=json_default_numpy),
                self.result.get("worst_val"),
                self.result.get("avg"),
                self.result.get("early_stopped", 0),
                self.duration,
            )
        )
        print(summary)
        self.log(summary)
        self._log_fd.close()
        return self.result

    def experiment_json(self):
        d = {
            "project": "maggy_amd",
            "user": None,
            "name": self.name,
            "module": "maggy",
            "app_id": str(self.app_id),
            "start": time.strftime("%Y-%m-%dT%H:%M:%S",
                                   time.localtime(self.job_start)),
            "executors": self.num_workers,
            "logdir": self.log_dir,
            "description": self.description,
            "experiment_type": self.controller.name(),
            "controller": self.controller.name(),
            "config": json.dumps(self.config_to_dict(),
                                 default=json_default_numpy),
        }
        if self.experiment_done:
            d["status"] = "FINISHED"
            d["finished"] = time.strftime(
                "%Y-%m-%dT%H:%M:%S", time.localtime(self.job_end))
            d["duration"] = self.duration
            d["config"] = json.dumps(self.result.get("best_config", {}),
                                     default=json_default_numpy)
            d["metric"] = self.result.get("best_val")
        else:
            d["status"] = "RUNNING"
        return json.dumps(d, default=json_default_numpy)

    def config_to_dict(self):
        return self.searchspace.to_dict() if self.searchspace else {}
