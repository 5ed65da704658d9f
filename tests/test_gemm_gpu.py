"""Numerics of the MFMA GEMM path vs fp32 torch references (GPU).

Every check compares the bf16 HIP kernel against the same computation in
fp32 torch on bf16-rounded inputs, with tolerances sized for bf16
accumulate-in-fp32 GEMM error.
"""
import pytest
import torch

pytestmark = pytest.mark.gpu

requires_gpu = pytest.mark.skipif(
    not torch.cuda.is_available(), reason="needs a GPU")


def _ext():
    from maggy_amd import ops

    return ops.require_ext()


def _rand(shape, seed):
    g = torch.Generator(device="cuda").manual_seed(seed)
    return (torch.rand(shape, generator=g, device="cuda") * 2 - 1).to(
        torch.bfloat16)


def _close(got, ref, K):
    # bf16 inputs, fp32 accumulate: error ~ sqrt(K) * 2^-8 relative
    tol = 3e-2 + 1.2e-3 * (K ** 0.5)
    err = (got.float() - ref).abs()
    scale = ref.abs().mean().clamp(min=1.0)
    rel = (err / (ref.abs() + scale)).max().item()
    assert rel < tol, "max rel err {} > {}".format(rel, tol)


@requires_gpu
@pytest.mark.parametrize("M,N,K", [(256, 256, 64), (512, 256, 192),
                                   (512, 512, 512), (1024, 512, 2048)])
def test_gemm_tn_numerics(M, N, K):
    ext = _ext()
    a = _rand((M, K), 1)
    w = _rand((N, K), 2)
    got = ext.gemm_tn(a, w)
    ref = a.float() @ w.float().t()
    _close(got, ref, K)


@requires_gpu
def test_transpose2d():
    ext = _ext()
    x = _rand((192, 448), 3)
    xt = ext.transpose2d(x)
    assert torch.equal(xt, x.t().contiguous())


@requires_gpu
def test_gemm_tn_swiglu_fused():
    ext = _ext()
    M, N, K = 512, 256, 256
    a = _rand((M, K), 4)
    w1 = _rand((N, K), 5)
    w3 = _rand((N, K), 6)
    y1 = ext.gemm_tn(a, w1)
    y3, h = ext.gemm_tn_swiglu(a, w3, y1)
    ref_y3 = a.float() @ w3.float().t()
    _close(y3, ref_y3, K)
    ref_h = torch.nn.functional.silu(y1.float()) * ref_y3
    _close(h, ref_h, K)


@requires_gpu
def test_maggy_linear_fwd_bwd_vs_torch():
    """Full autograd round trip of maggy_linear vs F.linear in fp32."""
    from maggy_amd.ops.linear import maggy_linear

    M, N, K = 512, 512, 256
    x = _rand((M, K), 7).requires_grad_(True)
    w = _rand((N, K), 8).requires_grad_(True)
    y = maggy_linear(x, w)
    dy = _rand((M, N), 9)
    y.backward(dy)

    xr = x.detach().float().clone().requires_grad_(True)
    wr = w.detach().float().clone().requires_grad_(True)
    yr = torch.nn.functional.linear(xr, wr)
    yr.backward(dy.float())

    _close(y, yr.detach(), K)
    _close(x.grad, xr.grad, N)
    _close(w.grad, wr.grad, M)


@requires_gpu
def test_maggy_feedforward_vs_eager():
    """Fused-MLP module (custom GEMM + SwiGLU epilogue) against the eager
    fp32 computation, forward and backward."""
    from maggy_amd.ops.linear import MaggyFeedForward

    torch.manual_seed(0)
    dim, hidden, M = 256, 512, 512
    ff = MaggyFeedForward(dim, hidden).cuda().to(torch.bfloat16)
    x = _rand((2, M // 2, dim), 10).requires_grad_(True)
    out = ff(x)
    assert out.shape == x.shape
    loss = out.float().square().mean()
    loss.backward()

    xr = x.detach().float().clone().requires_grad_(True)
    w1 = ff.w1.weight.detach().float()
    w3 = ff.w3.weight.detach().float()
    w2 = ff.w2.weight.detach().float()
    h = torch.nn.functional.silu(xr @ w1.t()) * (xr @ w3.t())
    outr = h @ w2.t()
    lossr = outr.square().mean()
    lossr.backward()

    _close(out.reshape(-1, dim), outr.detach().reshape(-1, dim), hidden)
    rel = (x.grad.float() - xr.grad).abs().max() / \
        (xr.grad.abs().max() + 1e-6)
    assert rel < 0.1, rel
    assert ff.w1.weight.grad is not None
    assert ff.w2.weight.grad is not None


@requires_gpu
def test_maggy_linear_fallback_shapes():
    """Non-tiling shapes fall back to F.linear (still correct)."""
    from maggy_amd.ops.linear import MaggyLinear

    lin = MaggyLinear(100, 60).cuda().to(torch.bfloat16)
    x = _rand((7, 100), 11)
    y = lin(x)
    ref = x.float() @ lin.weight.float().t()
    _close(y, ref, 100)
