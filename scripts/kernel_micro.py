"""Fixed-shape microbench driver for PMC profiling of the HIP kernels.

Runs each hand-written kernel a fixed number of times at a documented
shape so `rocprofv3 --pmc` rows can be matched to kernels and compared
against theoretical byte counts.

    python scripts/kernel_micro.py [--iters 20] [--only NAME]

Shapes (bf16 unless noted):
  rms      : [32768, 4096]             1 r + 1 w = 512 MiB/call/dir
  swiglu   : g,u [16384, 14336]        3 sweeps of 469 MiB
  bn       : NHWC [256, 56, 56, 64]    fwd reduce+apply
  adam     : 64 MiB params (fp32 master + bf16 grads)
  reduce   : 256 Mi elements bf16 sum
  gemm     : 4096^3 TN
  transpose: [16384, 4096]
"""
import argparse
import os
import sys

import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

from maggy_amd import ops  # noqa: E402


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--iters", type=int, default=20)
    ap.add_argument("--only", default="")
    args = ap.parse_args()
    ext = ops.require_ext()
    it = args.iters
    dev = "cuda"

    def on(name):
        return not args.only or args.only == name

    if on("rms"):
        R, D = 32768, 4096
        x = torch.randn(R, D, device=dev).bfloat16()
        w = torch.ones(D, device=dev)
        y = torch.empty_like(x)
        inv = torch.empty(R, device=dev)
        for _ in range(it):
            ext.rms_fwd(x, w, y, inv, R, D, 1e-5)
        dy = torch.randn_like(x)
        dx = torch.empty_like(x)
        dwp = torch.empty(D * 1024, device=dev)
        dw = torch.empty(D, device=dev)
        for _ in range(it):
            ext.rms_bwd(dy, x, w, inv, dx, dwp, dw, R, D)

    if on("swiglu"):
        g = torch.randn(16384, 14336, device=dev).bfloat16()
        u = torch.randn_like(g)
        out = torch.empty_like(g)
        for _ in range(it):
            ext.swiglu_fwd(g, u, out)

    if on("adam"):
        from maggy_amd.ops.fused_adam import FusedAdam

        params = [torch.randn(1024, 4096, device=dev).bfloat16()
                  .requires_grad_(True) for _ in range(16)]
        for p in params:
            p.grad = torch.randn_like(p)
        opt = FusedAdam(params, lr=1e-3)
        for _ in range(it):
            opt.step()

    if on("bn"):
        from maggy_amd.ops.fused_bn import MaggyBatchNorm2d

        bn = MaggyBatchNorm2d(64).cuda().to(torch.bfloat16)
        x = torch.randn(256, 64, 56, 56, device=dev).bfloat16() \
            .to(memory_format=torch.channels_last).requires_grad_(True)
        for _ in range(it):
            y = bn(x)
            y.backward(torch.randn_like(y))
            x.grad = None

    if on("reduce"):
        x = torch.randn(256 * 1024 * 1024, device=dev).bfloat16()
        out = torch.zeros(1, device=dev)
        for _ in range(it):
            ext.reduce_sum(x, out)

    if on("gemm"):
        a = (torch.rand(4096, 4096, device=dev) * 2 - 1).bfloat16()
        w = (torch.rand(4096, 4096, device=dev) * 2 - 1).bfloat16()
        for _ in range(it):
            ext.gemm_tn(a, w)

    if on("rope"):
        from maggy_amd.ops.fused_rms import rope_bthd
        from maggy_amd.models.llama import precompute_rope

        cos, sin = precompute_rope(128, 4096, 500000.0)
        cos, sin = cos.cuda(), sin.cuda()
        x = torch.randn(4, 4096, 32, 128, device=dev).bfloat16()
        for _ in range(it):
            rope_bthd(x, cos, sin, 0)

    if on("transpose"):
        x = torch.randn(16384, 4096, device=dev).bfloat16()
        for _ in range(it):
            ext.transpose2d(x)

    torch.cuda.synchronize()
    print("kernel_micro done")


if __name__ == "__main__":
    main()
