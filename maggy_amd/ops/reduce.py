"""Scalar reductions on the HIP reduction kernels.

The per-trial metric reduction behind ``reporter.broadcast`` (reference
call-site N8): a training function can hand broadcast() a
GPU tensor (per-sample losses/accuracies) and the mean is computed by the
hierarchical wave->LDS->atomic HIP kernel, not by a torch eager reduction.
"""
import torch

from maggy_amd import ops


def metric_sum(t):
    """Sum of a CUDA fp32/bf16 tensor -> python float (HIP kernel)."""
    if not t.is_cuda:
        return float(t.float().sum())
    ext = ops.require_ext()
    out = torch.zeros(1, dtype=torch.float32, device=t.device)
    ext.reduce_sum(t.contiguous(), out)
    return float(out.item())


def metric_mean(t):
    if t.numel() == 0:
        return 0.0
    return metric_sum(t) / t.numel()


def metric_max(t):
    if not t.is_cuda:
        return float(t.float().max())
    ext = ops.require_ext()
    out = torch.full((1,), float("-inf"), dtype=torch.float32,
                     device=t.device)
    ext.reduce_max(t.contiguous().float(), out)
    return float(out.item())


def grad_l2norm(params):
    """Global L2 norm over the .grad tensors of ``params`` (HIP kernel on
    GPU, torch fallback on CPU)."""
    grads = [p.grad for p in params if p.grad is not None]
    if not grads:
        return 0.0
    if not grads[0].is_cuda:
        total = sum(float(g.float().pow(2).sum()) for g in grads)
        return total ** 0.5
    from maggy_amd.ops.fused_adam import _pack_tables

    ext = ops.require_ext()
    device = grads[0].device
    bf16 = [g for g in grads if g.dtype == torch.bfloat16]
    f32 = [g for g in grads if g.dtype == torch.float32]
    out = torch.zeros(1, dtype=torch.float32, device=device)
    for group, is_bf16 in ((bf16, True), (f32, False)):
        if not group:
            continue
        metas = [(g.data_ptr(), 0, g.data_ptr(), g.data_ptr(), g.data_ptr(),
                  g.numel()) for g in group]
        t, c, n = _pack_tables(metas, device)
        ext.multi_l2norm_sq(c, n, t, is_bf16, out)
    return float(out.item()) ** 0.5
