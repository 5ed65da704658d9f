"""HIP kernel numerics vs plain PyTorch fp32 references (GPU only)."""
import pytest
import torch

pytestmark = pytest.mark.gpu

if not torch.cuda.is_available():
    pytest.skip("needs an AMD GPU", allow_module_level=True)


@pytest.fixture(scope="module")
def ext():
    from maggy_amd import ops

    return ops.require_ext()


def test_reduce_sum_f32(ext):
    from maggy_amd.ops import metric_sum

    x = torch.randn(1_000_003, device="cuda")
    assert abs(metric_sum(x) - float(x.double().sum())) < 1e-1


def test_reduce_sum_bf16(ext):
    from maggy_amd.ops import metric_mean

    x = torch.randn(65_536, device="cuda").bfloat16()
    ref = float(x.float().mean())
    assert abs(metric_mean(x) - ref) < 1e-3


def test_reduce_max(ext):
    from maggy_amd.ops.reduce import metric_max

    x = torch.randn(123_457, device="cuda")
    assert metric_max(x) == pytest.approx(float(x.max()), rel=1e-6)
    x = -x.abs()  # all negative
    assert metric_max(x) == pytest.approx(float(x.max()), rel=1e-6)


def test_grad_l2norm(ext):
    from maggy_amd.ops import grad_l2norm

    ps = [torch.nn.Parameter(torch.randn(n, device="cuda"))
          for n in (1000, 32768, 70_001)]
    for p in ps:
        p.grad = torch.randn_like(p)
    ref = torch.sqrt(sum(p.grad.pow(2).sum() for p in ps)).item()
    assert grad_l2norm(ps) == pytest.approx(ref, rel=1e-5)


def _ref_adamw(params_ref, grads, lr, b1, b2, eps, wd, steps):
    """Plain fp32 torch AdamW reference (same math as the kernel)."""
    ms = [torch.zeros_like(p) for p in params_ref]
    vs = [torch.zeros_like(p) for p in params_ref]
    for t in range(1, steps + 1):
        bc1 = 1 - b1 ** t
        bc2 = 1 - b2 ** t
        for p, g, m, v in zip(params_ref, grads[t - 1], ms, vs):
            p.mul_(1 - lr * wd)
            m.mul_(b1).add_(g, alpha=1 - b1)
            v.mul_(b2).addcmul_(g, g, value=1 - b2)
            denom = (v.sqrt() / (bc2 ** 0.5)).add_(eps)
            p.addcdiv_(m, denom, value=-lr / bc1)
    return params_ref


def test_fused_adam_fp32_matches_reference(ext):
    from maggy_amd.ops import FusedAdam

    torch.manual_seed(0)
    shapes = [(1000,), (300, 200), (7,), (64, 3, 3, 3)]
    params = [torch.nn.Parameter(torch.randn(s, device="cuda"))
              for s in shapes]
    ref = [p.detach().clone() for p in params]
    opt = FusedAdam(params, lr=1e-2, betas=(0.9, 0.999), eps=1e-8,
                    weight_decay=0.01)
    steps = 5
    all_grads = []
    for _ in range(steps):
        gs = [torch.randn_like(p) for p in params]
        all_grads.append(gs)
        for p, g in zip(params, gs):
            p.grad = g.clone()
        opt.step()
    _ref_adamw(ref, all_grads, 1e-2, 0.9, 0.999, 1e-8, 0.01, steps)
    for p, r in zip(params, ref):
        torch.testing.assert_close(p.detach(), r, rtol=2e-5, atol=2e-6)


def test_fused_adam_bf16_master_weights(ext):
    from maggy_amd.ops import FusedAdam

    torch.manual_seed(1)
    p32 = torch.randn(4096, device="cuda")
    p_bf = torch.nn.Parameter(p32.bfloat16())
    ref = p_bf.detach().float().clone()
    opt = FusedAdam([p_bf], lr=1e-2)
    all_grads = []
    for _ in range(3):
        g = torch.randn(4096, device="cuda").bfloat16()
        all_grads.append([g.float()])
        p_bf.grad = g
        opt.step()
    _ref_adamw([ref], all_grads, 1e-2, 0.9, 0.999, 1e-8, 0.0, 3)
    # master weights must match the fp32 reference closely
    master = opt.state[p_bf]["master"]
    torch.testing.assert_close(master, ref, rtol=2e-5, atol=2e-6)
    # the bf16 mirror is the rounded master
    torch.testing.assert_close(p_bf.detach(), ref.bfloat16(),
                               rtol=1e-2, atol=1e-2)


def test_fused_adam_grad_clip(ext):
    from maggy_amd.ops import FusedAdam

    torch.manual_seed(2)
    p = torch.nn.Parameter(torch.zeros(1000, device="cuda"))
    g = torch.randn(1000, device="cuda") * 100.0  # huge grads
    ref = p.detach().clone()
    opt = FusedAdam([p], lr=1e-2, max_grad_norm=1.0)
    p.grad = g.clone()
    opt.step()
    nrm = g.norm()
    clipped = [g * (1.0 / (nrm + 1e-6))]
    _ref_adamw([ref], [clipped], 1e-2, 0.9, 0.999, 1e-8, 0.0, 1)
    torch.testing.assert_close(p.detach(), ref, rtol=1e-4, atol=1e-6)


def test_fused_sgd_momentum_matches_torch(ext):
    from maggy_amd.ops import FusedSGD

    torch.manual_seed(3)
    shapes = [(513,), (32, 17)]
    params = [torch.nn.Parameter(torch.randn(s, device="cuda"))
              for s in shapes]
    ref = [torch.nn.Parameter(p.detach().clone()) for p in params]
    opt = FusedSGD(params, lr=0.1, momentum=0.9, weight_decay=1e-4)
    topt = torch.optim.SGD(ref, lr=0.1, momentum=0.9, weight_decay=1e-4)
    for _ in range(4):
        for p, r in zip(params, ref):
            g = torch.randn_like(p)
            p.grad = g.clone()
            r.grad = g.clone()
        opt.step()
        topt.step()
    for p, r in zip(params, ref):
        torch.testing.assert_close(p.detach(), r.detach(), rtol=2e-5,
                                   atol=2e-6)


def test_fused_adam_trains_a_model(ext):
    """Loss must decrease on a real (tiny) model under autocast bf16."""
    from maggy_amd.models import resnet18_thin
    from maggy_amd.ops import FusedAdam

    torch.manual_seed(0)
    model = resnet18_thin(num_classes=10).cuda()
    opt = FusedAdam(model.parameters(), lr=1e-3, max_grad_norm=5.0)
    x = torch.randn(16, 3, 32, 32, device="cuda")
    y = torch.randint(0, 10, (16,), device="cuda")
    losses = []
    for _ in range(20):
        opt.zero_grad(set_to_none=True)
        with torch.autocast("cuda", dtype=torch.bfloat16):
            loss = torch.nn.functional.cross_entropy(model(x), y)
        loss.backward()
        opt.step()
        losses.append(float(loss.detach()))
    assert losses[-1] < losses[0] * 0.5, losses
