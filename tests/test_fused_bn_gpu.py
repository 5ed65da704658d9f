"""Fused BN kernel numerics vs plain fp32 torch BatchNorm (GPU only)."""
import pytest
import torch
import torch.nn.functional as F

pytestmark = pytest.mark.gpu

if not torch.cuda.is_available():
    pytest.skip("needs an AMD GPU", allow_module_level=True)


def _mk(n, c, h, w, seed=0):
    torch.manual_seed(seed)
    x = torch.randn(n, c, h, w, device="cuda").bfloat16().to(
        memory_format=torch.channels_last)
    return x


def _ref_bn(x32, weight, bias, relu=False, residual=None, eps=1e-5):
    ref = F.batch_norm(x32, None, None, weight, bias, True, 0.1, eps)
    if residual is not None:
        ref = ref + residual
    if relu:
        ref = F.relu(ref)
    return ref


@pytest.mark.parametrize("C,relu", [(64, False), (64, True), (128, True),
                                    (256, True)])
def test_bn_forward_matches_fp32(C, relu):
    from maggy_amd.ops.fused_bn import MaggyBatchNorm2d

    bn = MaggyBatchNorm2d(C, relu=relu).cuda()
    x = _mk(8, C, 14, 14)
    y = bn(x)
    ref = _ref_bn(x.float(), bn.weight, bn.bias, relu=relu)
    assert y.dtype == torch.bfloat16
    torch.testing.assert_close(y.float(), ref, rtol=2e-2, atol=2e-2)
    # running stats updated toward batch stats
    batch_mean = x.float().mean(dim=(0, 2, 3))
    torch.testing.assert_close(bn.running_mean, 0.1 * batch_mean,
                               rtol=1e-2, atol=1e-2)


def test_bn_forward_with_residual():
    from maggy_amd.ops.fused_bn import MaggyBatchNorm2d

    C = 64
    bn = MaggyBatchNorm2d(C, relu=True).cuda()
    x = _mk(4, C, 8, 8, seed=1)
    res = _mk(4, C, 8, 8, seed=2)
    y = bn(x, residual=res)
    ref = _ref_bn(x.float(), bn.weight, bn.bias, relu=True,
                  residual=res.float())
    torch.testing.assert_close(y.float(), ref, rtol=2e-2, atol=2e-2)


@pytest.mark.parametrize("relu", [False, True])
def test_bn_backward_matches_fp32(relu):
    from maggy_amd.ops.fused_bn import MaggyBatchNorm2d

    C = 64
    bn = MaggyBatchNorm2d(C, relu=relu).cuda()
    x = _mk(8, C, 10, 10, seed=3)
    x_fused = x.detach().clone().requires_grad_(True)
    y = bn(x_fused)
    dy = torch.randn_like(y)
    y.backward(dy)

    x_ref = x.detach().float().clone().requires_grad_(True)
    w_ref = bn.weight.detach().clone().requires_grad_(True)
    b_ref = bn.bias.detach().clone().requires_grad_(True)
    ref = _ref_bn(x_ref, w_ref, b_ref, relu=relu)
    ref.backward(dy.float())

    torch.testing.assert_close(x_fused.grad.float(), x_ref.grad, rtol=5e-2,
                               atol=5e-2)
    torch.testing.assert_close(bn.weight.grad, w_ref.grad, rtol=2e-2,
                               atol=2e-1)
    torch.testing.assert_close(bn.bias.grad, b_ref.grad, rtol=2e-2,
                               atol=2e-1)


def test_bn_backward_with_residual_grad():
    from maggy_amd.ops.fused_bn import MaggyBatchNorm2d

    C = 64
    bn = MaggyBatchNorm2d(C, relu=True).cuda()
    x = _mk(4, C, 8, 8, seed=4).requires_grad_(True)
    res = _mk(4, C, 8, 8, seed=5).requires_grad_(True)
    y = bn(x, residual=res)
    dy = torch.randn_like(y)
    y.backward(dy)

    x_ref = x.detach().float().clone().requires_grad_(True)
    r_ref = res.detach().float().clone().requires_grad_(True)
    w_ref = bn.weight.detach().clone().requires_grad_(True)
    b_ref = bn.bias.detach().clone().requires_grad_(True)
    ref = _ref_bn(x_ref, w_ref, b_ref, relu=True, residual=r_ref)
    ref.backward(dy.float())

    torch.testing.assert_close(res.grad.float(), r_ref.grad, rtol=5e-2,
                               atol=5e-2)
    torch.testing.assert_close(x.grad.float(), x_ref.grad, rtol=5e-2,
                               atol=5e-2)


def test_resnet_block_trains_with_fused_bn():
    from maggy_amd.models import resnet18_thin
    from maggy_amd.ops import FusedSGD

    torch.manual_seed(0)
    model = resnet18_thin(num_classes=10).cuda().to(
        memory_format=torch.channels_last)
    opt = FusedSGD(model.parameters(), lr=0.05, momentum=0.9)
    x = torch.randn(16, 3, 32, 32, device="cuda").to(
        memory_format=torch.channels_last)
    y = torch.randint(0, 10, (16,), device="cuda")
    losses = []
    for _ in range(25):
        opt.zero_grad(set_to_none=True)
        with torch.autocast("cuda", dtype=torch.bfloat16):
            loss = torch.nn.functional.cross_entropy(model(x), y)
        loss.backward()
        opt.step()
        losses.append(float(loss.detach()))
    assert losses[-1] < losses[0] * 0.6, losses


def test_bn_eval_mode():
    from maggy_amd.ops.fused_bn import MaggyBatchNorm2d

    C = 64
    bn = MaggyBatchNorm2d(C, relu=False).cuda()
    x = _mk(4, C, 8, 8, seed=6)
    bn(x)  # one training pass to move running stats
    bn.eval()
    y = bn(x)
    ref = F.batch_norm(x.float(), bn.running_mean, bn.running_var,
                       bn.weight, bn.bias, False, 0.1, bn.eps)
    torch.testing.assert_close(y.float(), ref, rtol=2e-2, atol=2e-2)
