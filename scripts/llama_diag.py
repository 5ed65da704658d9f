"""Diagnose Llama step cost on MI355X: sdpa backend, per-op timings."""
import time

import torch
import torch.nn.functional as F


def t(fn, iters=20, warmup=5):
    for _ in range(warmup):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters * 1e3


def main():
    dev = "cuda"
    B, T, D, H, KV, HD, FF, V = 8, 4096, 2048, 32, 8, 64, 8192, 128256
    print("flash_sdp:", torch.backends.cuda.flash_sdp_enabled(),
          "mem_eff:", torch.backends.cuda.mem_efficient_sdp_enabled(),
          "math:", torch.backends.cuda.math_sdp_enabled())

    q = torch.randn(B, H, T, HD, device=dev, dtype=torch.bfloat16)
    k = torch.randn(B, KV, T, HD, device=dev, dtype=torch.bfloat16)
    v = torch.randn(B, KV, T, HD, device=dev, dtype=torch.bfloat16)
    from torch.nn.attention import SDPBackend, sdpa_kernel

    for name, backend in [("FLASH", SDPBackend.FLASH_ATTENTION),
                          ("EFFICIENT", SDPBackend.EFFICIENT_ATTENTION),
                          ("MATH", SDPBackend.MATH)]:
        try:
            with sdpa_kernel(backend):
                ms = t(lambda: F.scaled_dot_product_attention(
                    q, k, v, is_causal=True, enable_gqa=True))
            print("sdpa", name, "fwd: {:.3f} ms".format(ms))
        except Exception as e:
            print("sdpa", name, "unavailable:", str(e)[:80])

    # default path fwd+bwd
    qg = q.clone().requires_grad_(True)
    dy = torch.randn(B, H, T, HD, device=dev, dtype=torch.bfloat16)

    def fb():
        out = F.scaled_dot_product_attention(qg, k, v, is_causal=True,
                                             enable_gqa=True)
        out.backward(dy)
        qg.grad = None

    print("sdpa default f+b: {:.3f} ms".format(t(fb)))

    # RMSNorm eager (the models/llama.py implementation)
    x = torch.randn(B, T, D, device=dev, dtype=torch.bfloat16)
    w = torch.ones(D, device=dev)

    def rms():
        xf = x.float()
        y = xf * torch.rsqrt(xf.pow(2).mean(-1, keepdim=True) + 1e-5)
        return (y * w).to(torch.bfloat16)

    print("rmsnorm eager fwd: {:.3f} ms  ({} MB tensor)".format(
        t(rms), x.numel() * 2 // 2**20))
    xg = x.clone().requires_grad_(True)
    wg = w.clone().requires_grad_(True)
    dyr = torch.randn_like(x)

    def rms_fb():
        xf = xg.float()
        y = xf * torch.rsqrt(xf.pow(2).mean(-1, keepdim=True) + 1e-5)
        (y * wg).to(torch.bfloat16).backward(dyr)
        xg.grad = None
        wg.grad = None

    print("rmsnorm eager f+b: {:.3f} ms".format(t(rms_fb)))

    # GEMM rates (hipblaslt): qkv-ish and ffn-ish
    a = torch.randn(B * T, D, device=dev, dtype=torch.bfloat16)
    wq = torch.randn(D, D, device=dev, dtype=torch.bfloat16)
    wf = torch.randn(D, FF, device=dev, dtype=torch.bfloat16)
    wv = torch.randn(D, V, device=dev, dtype=torch.bfloat16)
    for nm, ww in [("DxD", wq), ("DxFF", wf), ("DxV", wv)]:
        ms = t(lambda: a @ ww)
        fl = 2 * a.shape[0] * ww.shape[0] * ww.shape[1] / (ms / 1e3) / 1e12
        print("gemm {}: {:.3f} ms = {:.0f} TF/s".format(nm, ms, fl))

    # swiglu eager
    g = torch.randn(B * T, FF, device=dev, dtype=torch.bfloat16)
    u = torch.randn(B * T, FF, device=dev, dtype=torch.bfloat16)
    print("silu*mul eager: {:.3f} ms".format(t(lambda: F.silu(g) * u)))

    # cross entropy with fp32 logits materialization
    logits = torch.randn(B * T // 4, V, device=dev, dtype=torch.bfloat16)
    tgt = torch.randint(0, V, (B * T // 4,), device=dev)
    print("CE (.float()): {:.3f} ms".format(
        t(lambda: F.cross_entropy(logits.float(), tgt))))
    print("CE (bf16 in): {:.3f} ms".format(
        t(lambda: F.cross_entropy(logits, tgt))))


if __name__ == "__main__":
    main()
