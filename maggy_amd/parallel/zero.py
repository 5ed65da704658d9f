"""ZeRO-1 sharded fused optimizer.

Replaces the reference's 11 ``MaggyZero*`` wrappers around
torch.distributed.optim.ZeroRedundancyOptimizer
(/root/reference/maggy/core/patching/optim.py:28-117, call-site N3): each
rank owns a balanced shard of the parameters, updates it with the fused HIP
Adam/SGD kernel in one launch, and the updated shards are exchanged with
coalesced broadcasts over xGMI.
"""
import torch
import torch.distributed as dist

from maggy_amd.ops.fused_adam import FusedAdam, FusedSGD

# coalesced-broadcast bucket: xGMI links move ~153 GB/s; 128 MB buckets keep
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


class _ZeroShardedBase(torch.optim.Optimizer):
    inner_cls = None

    def __init__(self, params, process_group=None, **kwargs):
        params = [p for p in params if p.requires_grad]
        super().__init__(params, dict())
        self.pg = process_group
        self.world = dist.get_world_size(self.pg)
        self.rank = dist.get_rank(self.pg)
        all_params = [p for g in self.param_groups for p in g["params"]]
        self._all_params = all_params
        shards, owner = _partition(all_params, self.world)
        self._owner = owner
        self._my_shard = shards[self.rank]
        self.inner = self.inner_cls(self._my_shard, **kwargs) \
            if self._my_shard else None

    @torch.no_grad()
    def step(self, closure=None):
        loss = closure() if closure is not None else None
        if self.inner is not None:
            self.inner.step()
        # exchange updated shards: coalesced broadcast from each owner
        bufs = []
        for i, p in enumerate(self._all_params):
            bufs.append((self._owner[i], p.data))
        from torch.distributed import _broadcast_coalesced

        by_owner = {}
        for r, t in bufs:
            by_owner.setdefault(r, []).append(t)
        for r, tensors in sorted(by_owner.items()):
            _broadcast_coalesced(
                self.pg or dist.group.WORLD, tensors, BCAST_BUCKET_BYTES, r)
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
