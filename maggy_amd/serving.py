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
