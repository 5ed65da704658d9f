"""Fused BatchNorm2d(+ReLU)(+residual) module on the HIP kernels.

Drop-in replacement for the ``BN -> (add) -> ReLU`` pattern in channels_last
bf16 training on MI355X.  Collapses MIOpen's 4-kernel spatial BN plus the
eager add/relu glue (53% of a ResNet-50 bf16 step, profiles/r01) into 3
forward + 3 backward launches, bf16 in/out (half the traffic of autocast's
fp32 BN path), exact fp32 statistics.

Falls back to torch native batch_norm when: no CUDA, input not bf16
channels-last 4D, or C % 8 != 0.
"""
import torch
import torch.nn as nn
import torch.nn.functional as F

from maggy_amd import ops


class _FusedBNFunction(torch.autograd.Function):
    BN_NB = 1024  # partial rows, must match BN_NB in fused_bn.hip

    @staticmethod
    def forward(ctx, x, residual, weight, bias, running_mean, running_var,
                training, momentum, eps, relu):
        ext = ops.require_ext()
        N, C, H, W = x.shape
        M = N * H * W
        y = torch.empty_like(x)
        ws = torch.empty(6 * C, dtype=torch.float32, device=x.device)
        partials = None
        if training:
            partials = torch.empty(2 * C * _FusedBNFunction.BN_NB,
                                   dtype=torch.float32, device=x.device)
        ext.bn_fwd(x, residual, y, M, C, weight, bias, running_mean,
                   running_var, momentum, eps, training, relu, ws, partials)
        ctx.save_for_backward(x, y, weight, ws)
        ctx.bn_shape = (M, C)
        ctx.relu = relu
        ctx.has_residual = residual is not None
        return y

    @staticmethod
    def backward(ctx, dy):
        ext = ops.require_ext()
        x, y, weight, ws = ctx.saved_tensors
        M, C = ctx.bn_shape
        dy = dy.contiguous(memory_format=torch.channels_last)
        dx = torch.empty_like(dy)
        dres = torch.empty_like(dy) if ctx.has_residual else None
        dgamma = torch.empty(C, dtype=torch.float32, device=dy.device)
        dbeta = torch.empty(C, dtype=torch.float32, device=dy.device)
        bwd_ws = torch.empty(5 * C, dtype=torch.float32, device=dy.device)
        partials = torch.empty(2 * C * _FusedBNFunction.BN_NB,
                               dtype=torch.float32, device=dy.device)
        ext.bn_bwd(dy, x, y, dx, dres, M, C, weight, ws, ctx.relu, dgamma,
                   dbeta, bwd_ws, partials)
        return (dx, dres, dgamma, dbeta, None, None, None, None, None, None)


class MaggyBatchNorm2d(nn.BatchNorm2d):
    """BatchNorm2d with optional fused ReLU and fused residual add.

    ``forward(x, residual=None)`` computes
    ``act(bn(x) + residual)`` where act is ReLU when ``relu=True``.
    """

    def __init__(self, num_features, eps=1e-5, momentum=0.1, relu=False):
        super().__init__(num_features, eps=eps, momentum=momentum,
                         affine=True, track_running_stats=True)
        self.relu = relu

    @staticmethod
    def _eligible_channels(C):
        # kernels require power-of-2 C in [8, 2048] (fixed-channel stride
        # and LDS fold contracts)
        return 8 <= C <= 2048 and (C & (C - 1)) == 0

    def _use_fused(self, x):
        return (
            x.is_cuda
            and x.dtype == torch.bfloat16
            and x.dim() == 4
            and self._eligible_channels(x.shape[1])
            and x.is_contiguous(memory_format=torch.channels_last)
            and ops.has_ext()
        )

    def forward(self, x, residual=None):
        if self._use_fused(x) and (
                residual is None
                or residual.is_contiguous(
                    memory_format=torch.channels_last)):
            return _FusedBNFunction.apply(
                x, residual, self.weight, self.bias, self.running_mean,
                self.running_var, self.training, self.momentum, self.eps,
                self.relu)
        if (x.is_cuda and x.dtype == torch.bfloat16 and x.dim() == 4
                and self._eligible_channels(x.shape[1])
                and not ops.has_ext()):
            # eligible input but extension missing: fail loudly, no silent
            # eager fallback on a GPU box (framework policy)
            ops.require_ext()
        out = F.batch_norm(
            x.float(), self.running_mean, self.running_var, self.weight,
            self.bias, self.training, self.momentum, self.eps)
        if residual is not None:
            out = out + residual.float()
        if self.relu:
            out = F.relu(out)
        return out.to(x.dtype)
