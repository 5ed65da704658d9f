"""ZeRO sharded fused optimizers.

Replaces the reference's 11 ``MaggyZero*`` wrappers around
torch.distributed.optim.ZeroRedundancyOptimizer
(/root/reference/maggy/core/patching/optim.py:28-117, call-site N3) and the
gradient/parameter sharding its DeepSpeed/fairscale backends provided
(/root/reference/maggy/core/patching/modules.py:68-139):

* level 1 (``grad_shard=False``): optimizer-STATE sharding.  Each rank owns
  a balanced shard of the parameters and updates it with the fused HIP
  Adam/SGD kernel in one launch; gradients are synchronized by DDP as
  usual; updated shards are exchanged with coalesced broadcasts over xGMI.
* level 2 (``grad_shard=True``): additionally shards the GRADIENT
  reduction — the module is NOT DDP-wrapped; ``step()`` reduces each
  owner's coalesced gradient bucket to that owner only (1/world of the
  reduction work and received bytes per rank), frees non-owned grads,
  steps the fused optimizer on the local shard, and broadcasts updated
  parameters.

Only public torch.distributed APIs are used (round-1 ADVICE: the private
``_broadcast_coalesced`` breaks across torch upgrades).
"""
import torch
import torch.distributed as dist

from maggy_amd.ops.fused_adam import FusedAdam, FusedSGD

# coalesced-exchange bucket: xGMI links move ~153 GB/s; 128 MB buckets keep
# per-bucket transfer time well above launch overhead
BCAST_BUCKET_BYTES = 128 * 1024 * 1024


def _partition(params, world_size):
    """Greedy balanced partition by numel; returns rank -> [param]."""
    order = sorted(range(len(params)), key=lambda i: -params[i].numel())
    loads = [0] * world_size
    shards = [[] for _ in range(world_size)]
    owner = {}
    for i in order:
        r = loads.index(min(loads))
        loads[r] += params[i].numel()
        shards[r].append(params[i])
        owner[i] = r
    return shards, owner


def _dtype_buckets(tensors, cap_bytes=BCAST_BUCKET_BYTES):
    """Split tensors into same-dtype/device groups bounded by cap_bytes."""
    buckets = []
    cur, cur_bytes, cur_key = [], 0, None
    for t in tensors:
        key = (t.dtype, t.device)
        nbytes = t.numel() * t.element_size()
        if cur and (key != cur_key or cur_bytes + nbytes > cap_bytes):
            buckets.append(cur)
            cur, cur_bytes = [], 0
        cur.append(t)
        cur_key = key
        cur_bytes += nbytes
    if cur:
        buckets.append(cur)
    return buckets


def _flatten(bucket):
    flat = torch.empty(sum(t.numel() for t in bucket), dtype=bucket[0].dtype,
                       device=bucket[0].device)
    off = 0
    for t in bucket:
        flat[off:off + t.numel()].copy_(t.reshape(-1))
        off += t.numel()
    return flat


def _unflatten_into(flat, bucket):
    off = 0
    for t in bucket:
        t.reshape(-1).copy_(flat[off:off + t.numel()])
        off += t.numel()


class _ZeroShardedBase(torch.optim.Optimizer):
    inner_cls = None

    def __init__(self, params, process_group=None, grad_shard=False,
                 **kwargs):
        params = [p for p in params if p.requires_grad]
        super().__init__(params, dict())
        self.pg = process_group
        self.world = dist.get_world_size(self.pg)
        self.rank = dist.get_rank(self.pg)
        self.grad_shard = grad_shard
        all_params = [p for g in self.param_groups for p in g["params"]]
        self._all_params = all_params
        shards, owner = _partition(all_params, self.world)
        self._owner = owner
        self._shards = shards
        self._my_shard = shards[self.rank]
        self.inner = self.inner_cls(self._my_shard, **kwargs) \
            if self._my_shard else None

    def _reduce_scatter_grads(self):
        """ZeRO-2 gradient exchange: each owner's coalesced grad bucket is
        summed to the owner only; non-owned grads are freed afterwards."""
        pg = self.pg or dist.group.WORLD
        for r, shard in enumerate(self._shards):
            grads = [p.grad for p in shard if p.grad is not None]
            if not grads:
                continue
            for bucket in _dtype_buckets(grads):
                flat = _flatten(bucket)
                dist.reduce(flat, dst=dist.get_global_rank(pg, r), group=pg)
                if r == self.rank:
                    flat.div_(self.world)
                    _unflatten_into(flat, bucket)
        # free grads this rank does not own (the memory win of ZeRO-2)
        for i, p in enumerate(self._all_params):
            if self._owner[i] != self.rank:
                p.grad = None

    def _broadcast_params(self):
        """Exchange updated shards: coalesced broadcast from each owner
        using the public API on flattened same-dtype buckets."""
        pg = self.pg or dist.group.WORLD
        for r, shard in enumerate(self._shards):
            if not shard:
                continue
            src = dist.get_global_rank(pg, r)
            for bucket in _dtype_buckets([p.data for p in shard]):
                flat = _flatten(bucket)
                dist.broadcast(flat, src=src, group=pg)
                if r != self.rank:
                    _unflatten_into(flat, bucket)

    @torch.no_grad()
    def step(self, closure=None):
        loss = closure() if closure is not None else None
        if self.grad_shard:
            self._reduce_scatter_grads()
        if self.inner is not None:
            self.inner.step()
        self._broadcast_params()
        return loss

    def zero_grad(self, set_to_none=True):
        for p in self._all_params:
            if p.grad is not None:
                if set_to_none:
                    p.grad = None
                else:
                    p.grad.zero_()


class ZeroFusedAdam(_ZeroShardedBase):
    inner_cls = FusedAdam


class ZeroFusedSGD(_ZeroShardedBase):
    inner_cls = FusedSGD


def broadcast_module_params(module, src=0, process_group=None):
    """One-time parameter/buffer sync at wrap time for the non-DDP ZeRO-2
    path (DDP normally does this in its ctor)."""
    pg = process_group or dist.group.WORLD
    tensors = [p.data for p in module.parameters()] + \
              [b.data for b in module.buffers()]
    for bucket in _dtype_buckets(tensors):
        flat = _flatten(bucket)
        dist.broadcast(flat, src=src, group=pg)
        if dist.get_rank(pg) != src:
            _unflatten_into(flat, bucket)
