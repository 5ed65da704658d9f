"""Minimal model serving: dynamic-batching inference engine.

The reference has no serving story; this rounds out the framework for
deployment use.  Core is framework-agnostic dynamic batching: requests
queue until ``max_batch`` items or ``max_wait_ms`` elapse, then one
batched forward runs (on the GPU when available) and each caller gets its
slice.  An optional HTTP surface (FastAPI, installed in this image) wraps
the same engine.

    server = ModelServer(predict_fn, max_batch=32, max_wait_ms=5)
    server.start()
    fut = server.submit(sample_tensor)      # thread-safe
    result = fut.result()

``predict_fn(batch)`` receives a stacked tensor (dim 0 = batch) and
returns a tensor with the same leading dimension.
"""
import queue
import threading
import time
from concurrent.futures import Future

import torch


class ModelServer:
    def __init__(self, predict_fn, max_batch=32, max_wait_ms=5.0,
                 device=None):
        self.predict_fn = predict_fn
        self.max_batch = max_batch
        self.max_wait_s = max_wait_ms / 1000.0
        if device is None:
            device = "cuda" if torch.cuda.is_available() else "cpu"
        self.device = torch.device(device)
        self._q = queue.Queue()
        self._thread = None
        self._stop = threading.Event()
        self.stats = {"requests": 0, "batches": 0}

    # -- lifecycle ------------------------------------------------------
    def start(self):
        if self._thread is not None:
            return self
        self._stop.clear()
        self._thread = threading.Thread(target=self._loop, daemon=True)
        self._thread.start()
        return self

    def stop(self):
        self._stop.set()
        if self._thread is not None:
            self._thread.join(timeout=10)
            self._thread = None

    def __enter__(self):
        return self.start()

    def __exit__(self, *exc):
        self.stop()

    # -- client API -----------------------------------------------------
    def submit(self, sample):
        """Queue one sample (tensor, unbatched); returns a Future."""
        fut = Future()
        self._q.put((sample, fut))
        return fut

    def predict(self, sample, timeout=30.0):
        return self.submit(sample).result(timeout=timeout)

    # -- batching loop ---------------------------------------------------
    def _loop(self):
        while not self._stop.is_set():
            try:
                first = self._q.get(timeout=0.05)
            except queue.Empty:
                continue
            batch = [first]
            deadline = time.perf_counter() + self.max_wait_s
            while len(batch) < self.max_batch:
                remaining = deadline - time.perf_counter()
                if remaining <= 0:
                    break
                try:
                    batch.append(self._q.get(timeout=remaining))
                except queue.Empty:
                    break
            samples = [b[0] for b in batch]
            futs = [b[1] for b in batch]
            try:
                stacked = torch.stack(
                    [torch.as_tensor(s) for s in samples]).to(self.device)
                with torch.no_grad():
                    out = self.predict_fn(stacked)
                out = out.cpu()
                for i, fut in enumerate(futs):
                    fut.set_result(out[i])
            except Exception as e:
                for fut in futs:
                    if not fut.done():
                        fut.set_exception(e)
            self.stats["requests"] += len(batch)
            self.stats["batches"] += 1


def _replica_main(replica_id, gpu_id, model_fn, req_q, resp_q, max_batch,
                  max_wait_s):
    """One serving replica: pin a GPU, build the model locally (nothing
    large crosses process boundaries — the trial-pool contract), run the
    dynamic-batching loop over the multiprocessing queues."""
    import os

    if gpu_id is not None:
        os.environ["HIP_VISIBLE_DEVICES"] = str(gpu_id)
        os.environ["CUDA_VISIBLE_DEVICES"] = str(gpu_id)
        os.environ.setdefault("HSA_ENABLE_IPC_MODE_LEGACY", "0")
    import torch as _torch

    predict_fn = model_fn()
    device = _torch.device(
        "cuda" if _torch.cuda.is_available() else "cpu")
    while True:
        first = req_q.get()
        if first is None:
            return
        batch = [first]
        deadline = time.perf_counter() + max_wait_s
        while len(batch) < max_batch:
            remaining = deadline - time.perf_counter()
            if remaining <= 0:
                break
            try:
                item = req_q.get(timeout=remaining)
            except queue.Empty:
                break
            if item is None:
                req_q.put(None)  # keep the shutdown signal visible
                break
            batch.append(item)
        rids = [b[0] for b in batch]
        try:
            stacked = _torch.stack(
                [_torch.as_tensor(b[1]) for b in batch]).to(device)
            with _torch.no_grad():
                out = predict_fn(stacked)
            out = out.cpu()
            for i, rid in enumerate(rids):
                resp_q.put((rid, True, out[i], replica_id))
        except Exception as e:  # report instead of dying
            for rid in rids:
                resp_q.put((rid, False, repr(e), replica_id))


class ReplicatedModelServer:
    """Multi-GPU serving: one replica process per GPU, requests routed to
    the least-loaded replica, dynamic batching inside each replica.

    ``model_fn`` is a picklable zero-arg callable executed INSIDE each
    replica after GPU pinning; it returns the ``predict_fn(batch)`` the
    replica serves (build the model there — the class-not-instance
    contract of the training engine applies to serving too).
    """

    def __init__(self, model_fn, n_replicas=None, gpu_ids=None,
                 max_batch=32, max_wait_ms=5.0, start_method="spawn"):
        import multiprocessing as mp

        from maggy_amd import util

        if n_replicas is None:
            n_replicas = max(1, util.num_gpus())
        if gpu_ids is None:
            n_gpu = util.num_gpus()
            gpu_ids = [i % n_gpu if n_gpu else None
                       for i in range(n_replicas)]
        self.n_replicas = n_replicas
        ctx = mp.get_context(start_method)
        self._resp_q = ctx.Queue()
        self._req_qs = [ctx.Queue() for _ in range(n_replicas)]
        self._procs = [
            ctx.Process(target=_replica_main,
                        args=(i, gpu_ids[i], model_fn, self._req_qs[i],
                              self._resp_q, max_batch, max_wait_ms / 1e3),
                        daemon=True)
            for i in range(n_replicas)
        ]
        self._futures = {}
        self._lock = threading.Lock()
        self._next_rid = 0
        self._outstanding = [0] * n_replicas
        self.stats = {"requests": 0,
                      "per_replica": [0] * n_replicas}
        self._collector = None
        self._stop = threading.Event()

    def start(self):
        if self._collector is not None:
            return self
        for p in self._procs:
            p.start()
        self._stop.clear()
        self._collector = threading.Thread(target=self._collect,
                                           daemon=True)
        self._collector.start()
        return self

    def _collect(self):
        while not self._stop.is_set():
            try:
                rid, ok, payload, replica = self._resp_q.get(timeout=0.1)
            except queue.Empty:
                continue
            with self._lock:
                fut = self._futures.pop(rid, None)
                self._outstanding[replica] -= 1
                self.stats["per_replica"][replica] += 1
            if fut is None:
                continue
            if ok:
                fut.set_result(payload)
            else:
                fut.set_exception(RuntimeError(payload))

    def submit(self, sample):
        fut = Future()
        with self._lock:
            rid = self._next_rid
            self._next_rid += 1
            replica = min(range(self.n_replicas),
                          key=lambda i: self._outstanding[i])
            self._outstanding[replica] += 1
            self._futures[rid] = fut
            self.stats["requests"] += 1
        self._req_qs[replica].put((rid, sample))
        return fut

    def predict(self, sample, timeout=30.0):
        return self.submit(sample).result(timeout=timeout)

    def stop(self):
        for q in self._req_qs:
            q.put(None)
        for p in self._procs:
            p.join(timeout=10)
            if p.is_alive():
                p.terminate()
        self._stop.set()
        if self._collector is not None:
            self._collector.join(timeout=5)
            self._collector = None

    def __enter__(self):
        return self.start()

    def __exit__(self, *exc):
        self.stop()


def make_app(server):
    """Optional FastAPI surface: POST /predict {"input": [...]} and
    GET /stats."""
    from fastapi import FastAPI
    from pydantic import BaseModel

    class PredictRequest(BaseModel):
        input: list

    app = FastAPI(title="maggy_amd model server")

    @app.post("/predict")
    def predict(req: PredictRequest):
        out = server.predict(torch.tensor(req.input))
        return {"output": out.tolist()}

    @app.get("/stats")
    def stats():
        return dict(server.stats)

    return app
