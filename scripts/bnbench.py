"""Microbench: fused BN kernels vs eager torch BN on ResNet-50 shapes."""
import time

import torch


def bench(fn, iters=30, warmup=5):
    for _ in range(warmup):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters * 1e3  # ms


def main():
    from maggy_amd.ops.fused_bn import MaggyBatchNorm2d

    N = 256
    shapes = [(64, 112, 112), (256, 56, 56), (512, 28, 28),
              (1024, 14, 14), (2048, 7, 7)]
    print(f"{'shape':>18} {'fwd ms':>8} {'GB/s':>7} {'f+b ms':>8} "
          f"{'GB/s':>7} {'eager f+b ms':>12}")
    for C, H, W in shapes:
        x = torch.randn(N, C, H, W, device="cuda").bfloat16().to(
            memory_format=torch.channels_last)
        bn = MaggyBatchNorm2d(C, relu=True).cuda()
        nbytes = x.numel() * 2

        fwd_ms = bench(lambda: bn(x))
        # fwd reads x twice (stats + apply) and writes y once
        fwd_gbs = 3 * nbytes / fwd_ms / 1e6

        xg = x.clone().requires_grad_(True)
        dy = torch.randn_like(x)

        def fb():
            y = bn(xg)
            y.backward(dy)
            xg.grad = None

        fb_ms = bench(fb)
        # + bwd: read dy,x,y twice (reduce+apply ~5 streams) write dx
        fb_gbs = (3 + 6) * nbytes / fb_ms / 1e6

        tbn = torch.nn.BatchNorm2d(C).cuda()
        xg2 = x.clone().requires_grad_(True)

        def eager():
            y = torch.relu(tbn(xg2.float()))
            y.backward(dy.float())
            xg2.grad = None

        eager_ms = bench(eager)
        print(f"{C:>5}x{H:>3}x{W:<3}      {fwd_ms:8.3f} {fwd_gbs:7.0f} "
              f"{fb_ms:8.3f} {fb_gbs:7.0f} {eager_ms:12.3f}")


if __name__ == "__main__":
    main()
