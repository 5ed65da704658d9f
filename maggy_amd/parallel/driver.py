"""Distributed-training experiment driver.

Parity: /root/reference/maggy/core/experiment_driver/
torch_distributed_training_driver.py:28-146 — runs the oblivious train_fn
as data-parallel training across all GPUs, final result = average of the
per-rank returned metrics.  The Spark fan-out + DistributedTrainingServer
are replaced by direct process spawning (one rank per GPU, RCCL rendezvous
on 127.0.0.1).
"""
import json
import multiprocessing as mp
import random
import time

from maggy_amd import util
from maggy_amd.constants import SCHEDULER
from maggy_amd.core import messages as M
from maggy_amd.core.environment import Environment
from maggy_amd.core.shm import MetricRing
from maggy_amd.exceptions import WorkerCrashError
from maggy_amd.parallel.worker import dist_worker_main
from maggy_amd.utils.jsonutil import json_default_numpy


class TorchDistributedTrainingDriver:
    def __init__(self, config, app_id=None, run_id=None):
        self.config = config
        self.name = config.name
        env = Environment.get_instance()
        self.app_id = app_id or env.get_app_id()
        self.run_id = run_id or env.next_run_id(self.app_id)
        self.log_dir = env.get_logdir(self.app_id, self.run_id)
        n_gpu = util.num_gpus()
        self.world_size = config.num_gpus or (n_gpu if n_gpu else 2)
        self.gpu_ids = [i % n_gpu if n_gpu else None
                        for i in range(self.world_size)]
        self.result = None
        self.job_start = None

    def run_experiment(self, train_fn):
        self.job_start = time.time()
        payload = {
            "train_fn": train_fn,
            "module": self.config.module,
            "dataset": self.config.dataset,
            "test_set": self.config.test_set,
            "hparams": self.config.hparams,
            "zero_lvl": self.config.zero_lvl,
            "mixed_precision": self.config.mixed_precision,
            "bucket_cap_mb": self.config.bucket_cap_mb,
            "master_addr": "127.0.0.1",
            "master_port": random.randint(20000, 49000),
        }
        ctx = mp.get_context("spawn")
        procs, conns, rings = [], [], []
        for rank in range(self.world_size):
            ring = MetricRing(slots=SCHEDULER.RING_SLOTS, create=True)
            parent, child = ctx.Pipe()
            p = ctx.Process(
                target=dist_worker_main,
                args=(rank, self.world_size, self.gpu_ids[rank], child,
                      ring.name, SCHEDULER.RING_SLOTS, self.log_dir,
                      payload),
                daemon=True,
            )
            p.start()
            child.close()
            procs.append(p)
            conns.append(parent)
            rings.append(ring)

        metrics = {}
        errors = {}
        metric_stream = {}
        try:
            pending = set(range(self.world_size))
            while pending:
                for rank in list(pending):
                    if conns[rank].poll(0.05):
                        msg = conns[rank].recv()
                        if msg[0] == M.FINAL:
                            metrics[rank] = msg[3]
                            pending.discard(rank)
                        elif msg[0] == M.ERROR:
                            errors[rank] = msg[3]
                            pending.discard(rank)
                    elif not procs[rank].is_alive():
                        errors[rank] = "rank {} died (exit {})".format(
                            rank, procs[rank].exitcode)
                        pending.discard(rank)
                for rank, ring in enumerate(rings):
                    for tag, step, value in ring.drain():
                        metric_stream.setdefault(rank, []).append(
                            (int(step), value))
        finally:
            for p in procs:
                p.join(timeout=SCHEDULER.JOIN_TIMEOUT)
                if p.is_alive():
                    p.terminate()
            for ring in rings:
                ring.close()
                ring.unlink()
            for c in conns:
                try:
                    c.close()
                except OSError:
                    pass

        if errors:
            raise WorkerCrashError(
                sorted(errors.keys()),
                "; ".join("rank {}: {}".format(r, e.splitlines()[-1]
                                               if isinstance(e, str) and e
                                               else e)
                          for r, e in sorted(errors.items())))

        # final = average of worker metrics (parity :138-146)
        vals = [v for v in metrics.values() if v is not None]
        avg = sum(vals) / len(vals) if vals else None
        self.result = {
            "final_metric_avg": avg,
            "per_rank": {str(r): v for r, v in sorted(metrics.items())},
            "world_size": self.world_size,
            "duration_ms": util.seconds_to_milliseconds(
                time.time() - self.job_start),
            # rank-0 heartbeat metric history (step, value)
            "metric_history": metric_stream.get(0, []),
        }
        Environment.get_instance().dump(
            json.dumps(self.result, default=json_default_numpy),
            self.log_dir + "/result.json",
        )
        return self.result
