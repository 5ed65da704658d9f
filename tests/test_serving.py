"""Dynamic-batching model server tests (CPU)."""
import threading

import pytest

import torch

from maggy_amd.models import MLP
from maggy_amd.serving import ModelServer, make_app


def test_batching_and_results():
    model = MLP(in_features=8, hidden=16, num_classes=3)
    model.eval()
    with ModelServer(model, max_batch=16, max_wait_ms=20) as server:
        xs = [torch.randn(8) for _ in range(40)]
        futs = [server.submit(x) for x in xs]
        outs = [f.result(timeout=10) for f in futs]
    with torch.no_grad():
        ref = model(torch.stack(xs))
    for o, r in zip(outs, ref):
        torch.testing.assert_close(o, r, rtol=1e-5, atol=1e-6)
    # batching actually happened (40 requests in < 40 batches)
    assert server.stats["requests"] == 40
    assert server.stats["batches"] < 40


def test_concurrent_clients():
    server = ModelServer(lambda b: b * 2.0, max_batch=8,
                         max_wait_ms=5).start()
    results = {}

    def client(i):
        results[i] = server.predict(torch.tensor([float(i)]))

    threads = [threading.Thread(target=client, args=(i,))
               for i in range(20)]
    for t in threads:
        t.start()
    for t in threads:
        t.join()
    server.stop()
    for i in range(20):
        assert float(results[i]) == 2.0 * i


def test_predict_fn_error_propagates():
    def boom(batch):
        raise RuntimeError("bad model")

    with ModelServer(boom, max_wait_ms=1) as server:
        fut = server.submit(torch.zeros(2))
        import pytest

        with pytest.raises(RuntimeError):
            fut.result(timeout=10)


def test_http_surface():
    from fastapi.testclient import TestClient

    with ModelServer(lambda b: b + 1.0, max_wait_ms=1) as server:
        app = make_app(server)
        client = TestClient(app)
        r = client.post("/predict", json={"input": [1.0, 2.0]})
        assert r.status_code == 200
        assert r.json()["output"] == [2.0, 3.0]
        assert client.get("/stats").json()["requests"] >= 1


def test_generation_serving():
    """Batched prompt-completion through the server (tiny Llama)."""
    from maggy_amd.models import LlamaConfig, LlamaModel

    torch.manual_seed(0)
    model = LlamaModel(LlamaConfig.tiny(vocab_size=50)).eval()

    def complete(batch):  # batch: [B, T] prompts -> [B, T+4] ids
        return model.generate(batch, max_new_tokens=4)

    with ModelServer(complete, max_batch=8, max_wait_ms=10) as server:
        prompts = [torch.randint(0, 50, (6,)) for _ in range(12)]
        outs = [server.submit(p).result(timeout=30) for p in prompts]
    for p, o in zip(prompts, outs):
        assert o.shape == (10,)
        assert torch.equal(o[:6], p)


def _replica_model_fn():
    """Builds the 'model' inside the replica (picklable, zero-arg)."""
    import torch

    def predict(batch):
        return batch * 2.0 + 1.0

    return predict


@pytest.mark.timeout(120)
def test_replicated_server_routes_and_serves():
    from maggy_amd.serving import ReplicatedModelServer

    with ReplicatedModelServer(_replica_model_fn, n_replicas=2,
                               gpu_ids=[None, None], max_batch=8,
                               max_wait_ms=10.0) as srv:
        futs = [srv.submit(torch.full((3,), float(i))) for i in range(24)]
        for i, f in enumerate(futs):
            out = f.result(timeout=30)
            assert torch.allclose(out, torch.full((3,), float(i) * 2 + 1))
        assert srv.stats["requests"] == 24
        # least-loaded routing actually used both replicas
        assert all(n > 0 for n in srv.stats["per_replica"]), \
            srv.stats["per_replica"]


@pytest.mark.timeout(120)
def test_replicated_server_error_propagates():
    from maggy_amd.serving import ReplicatedModelServer

    with ReplicatedModelServer(_failing_model_fn, n_replicas=1,
                               gpu_ids=[None], max_batch=4,
                               max_wait_ms=5.0) as srv:
        with pytest.raises(RuntimeError, match="boom"):
            srv.predict(torch.ones(2), timeout=30)
        # the replica survives a failing batch
        out = srv.predict(torch.zeros(2), timeout=30)
        assert torch.allclose(out, torch.zeros(2))


def _failing_model_fn():
    import torch

    def predict(batch):
        if batch.sum() > 0:
            raise ValueError("boom")
        return batch

    return predict


def _gpu_model_fn():
    import torch

    from maggy_amd.models import MLP

    torch.manual_seed(0)
    model = MLP(in_features=16, hidden=32, num_classes=4).cuda().eval()

    def predict(batch):
        return model(batch)

    return predict


@pytest.mark.gpu
@pytest.mark.timeout(180)
def test_replicated_server_gpu_replica():
    """One CUDA replica process serving a real model."""
    if not torch.cuda.is_available():
        pytest.skip("needs a GPU")
    from maggy_amd.serving import ReplicatedModelServer

    with ReplicatedModelServer(_gpu_model_fn, n_replicas=1, gpu_ids=[0],
                               max_batch=16, max_wait_ms=5.0) as srv:
        outs = [srv.predict(torch.randn(16), timeout=60) for _ in range(8)]
        assert all(o.shape == (4,) for o in outs)
        assert srv.stats["per_replica"][0] == 8


def _dying_model_fn():
    import os

    import torch  # noqa: F401

    def predict(batch):
        os._exit(17)  # hard replica death mid-batch

    return predict


@pytest.mark.timeout(120)
def test_replicated_server_replica_death_times_out_cleanly():
    """A replica that dies hard must not hang the client: the future
    times out and the server still shuts down."""
    from concurrent.futures import TimeoutError as FutTimeout

    from maggy_amd.serving import ReplicatedModelServer

    srv = ReplicatedModelServer(_dying_model_fn, n_replicas=1,
                                gpu_ids=[None], max_batch=4,
                                max_wait_ms=5.0).start()
    try:
        fut = srv.submit(torch.ones(2))
        with pytest.raises(FutTimeout):
            fut.result(timeout=3)
    finally:
        srv.stop()
