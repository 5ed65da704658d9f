"""Custom MFMA GEMM vs hipBLASLt (torch.matmul) on the Llama-3-8B
training shapes.  Run on a GPU box:

    python scripts/gemm_shapes_bench.py [--iters 20] > gpurun_out/gemm_shapes.txt

Shapes: M = tokens/step = 4 x 4096; the seven distinct linears of the 8B
config (q, kv, o, gate/up, down, lm_head).  For each: fwd (TN), dX
(TN on W^T image), dW (TN on transposed activations), with the transpose
cost counted against the custom path.
"""
import argparse
import os
import sys
import time

import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

from maggy_amd import ops  # noqa: E402


def bench(fn, iters=20, warmup=5):
    for _ in range(warmup):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters


def tf(M, N, K, sec):
    return 2.0 * M * N * K / sec / 1e12


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--iters", type=int, default=20)
    ap.add_argument("--m", type=int, default=16384)
    args = ap.parse_args()
    ext = ops.require_ext()
    M = args.m
    shapes = [
        ("q/o  ", M, 4096, 4096),
        ("kv   ", M, 1024, 4096),
        ("gate ", M, 14336, 4096),
        ("down ", M, 4096, 14336),
        ("head ", M, 128256, 4096),
    ]
    print("M=%d  iters=%d" % (M, args.iters))
    print("%-6s %-22s %9s %9s %9s" % ("shape", "op", "custom", "blaslt",
                                      "ratio"))
    for name, m, n, k in shapes:
        a = (torch.rand(m, k, device="cuda") * 2 - 1).bfloat16()
        w = (torch.rand(n, k, device="cuda") * 2 - 1).bfloat16()
        dy = (torch.rand(m, n, device="cuda") * 2 - 1).bfloat16()

        # fwd: Y = A @ W^T
        t_c = bench(lambda: ext.gemm_tn(a, w), args.iters)
        t_b = bench(lambda: a @ w.t(), args.iters)
        print("%-6s %-22s %7.0fTF %7.0fTF %8.2fx"
              % (name, "fwd  Y=X.W^T", tf(m, n, k, t_c), tf(m, n, k, t_b),
                 t_b / t_c))

        # dX = dY @ W  (custom: transpose W then TN; counted together)
        def dx_custom():
            return ext.gemm_tn(dy, ext.transpose2d(w))

        t_c = bench(dx_custom, args.iters)
        t_b = bench(lambda: dy @ w, args.iters)
        print("%-6s %-22s %7.0fTF %7.0fTF %8.2fx"
              % (name, "dX   dY.W (+Wt)", tf(m, n, k, t_c),
                 tf(m, n, k, t_b), t_b / t_c))

        # dW = dY^T @ X (custom: transpose both; counted together)
        def dw_custom():
            return ext.gemm_tn(ext.transpose2d(dy), ext.transpose2d(a))

        t_c = bench(dw_custom, args.iters)
        t_b = bench(lambda: dy.t() @ a, args.iters)
        print("%-6s %-22s %7.0fTF %7.0fTF %8.2fx"
              % (name, "dW   dY^T.X (+2T)", tf(m, n, k, t_c),
                 tf(m, n, k, t_b), t_b / t_c))
        del a, w, dy
        torch.cuda.empty_cache()

    # transpose bandwidth
    for r, c in [(16384, 4096), (16384, 14336), (4096, 16384)]:
        x = torch.randn(r, c, device="cuda").bfloat16()
        t = bench(lambda: ext.transpose2d(x), args.iters)
        gb = 2 * r * c * 2 / t / 1e9
        print("transpose %6dx%-6d %7.3f ms  %6.0f GB/s" % (r, c, t * 1e3,
                                                           gb))
        del x

    # fused swiglu GEMM vs separate
    m, n, k = M, 14336, 4096
    a = (torch.rand(m, k, device="cuda") * 2 - 1).bfloat16()
    w1 = (torch.rand(n, k, device="cuda") * 2 - 1).bfloat16()
    w3 = (torch.rand(n, k, device="cuda") * 2 - 1).bfloat16()
    y1 = ext.gemm_tn(a, w1)

    t_f = bench(lambda: ext.gemm_tn_swiglu(a, w3, y1), args.iters)

    def unfused():
        y3 = ext.gemm_tn(a, w3)
        out = torch.empty_like(y3)
        ext.swiglu_fwd(y1, y3, out)
        return out

    t_u = bench(unfused, args.iters)
    print("mlp swiglu: fused %7.3f ms  vs gemm+swiglu %7.3f ms  (%.2fx)"
          % (t_f * 1e3, t_u * 1e3, t_u / t_f))


if __name__ == "__main__":
    main()
