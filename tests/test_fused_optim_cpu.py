"""FusedAdam/FusedSGD CPU-path semantics (per-group hyperparams, clip)."""
import torch

from maggy_amd.ops import FusedAdam, FusedSGD


def test_adam_per_group_lr():
    a = torch.nn.Parameter(torch.ones(10))
    b = torch.nn.Parameter(torch.ones(10))
    opt = FusedAdam([
        {"params": [a], "lr": 1e-1},
        {"params": [b], "lr": 1e-3},
    ])
    a.grad = torch.ones(10)
    b.grad = torch.ones(10)
    opt.step()
    # both move opposite the gradient; the high-lr group moves further
    assert float(a.mean()) < float(b.mean()) < 1.0


def test_adam_matches_torch_adamw_cpu():
    torch.manual_seed(0)
    p1 = torch.nn.Parameter(torch.randn(64))
    p2 = torch.nn.Parameter(p1.detach().clone())
    opt = FusedAdam([p1], lr=1e-2, weight_decay=0.01)
    ref = torch.optim.AdamW([p2], lr=1e-2, weight_decay=0.01)
    for _ in range(5):
        g = torch.randn(64)
        p1.grad = g.clone()
        p2.grad = g.clone()
        opt.step()
        ref.step()
    torch.testing.assert_close(p1, p2, rtol=1e-5, atol=1e-6)


def test_sgd_matches_torch_cpu():
    torch.manual_seed(1)
    p1 = torch.nn.Parameter(torch.randn(32))
    p2 = torch.nn.Parameter(p1.detach().clone())
    opt = FusedSGD([p1], lr=0.1, momentum=0.9, weight_decay=1e-4)
    ref = torch.optim.SGD([p2], lr=0.1, momentum=0.9, weight_decay=1e-4)
    for _ in range(4):
        g = torch.randn(32)
        p1.grad = g.clone()
        p2.grad = g.clone()
        opt.step()
        ref.step()
    torch.testing.assert_close(p1, p2, rtol=1e-5, atol=1e-6)


def test_adam_grad_clip_cpu():
    p = torch.nn.Parameter(torch.zeros(100))
    p_ref = torch.nn.Parameter(torch.zeros(100))
    g = torch.randn(100) * 50
    opt = FusedAdam([p], lr=1e-2, max_grad_norm=1.0)
    ref = torch.optim.AdamW([p_ref], lr=1e-2, weight_decay=0.0)
    p.grad = g.clone()
    nrm = g.norm()
    p_ref.grad = g * (1.0 / (nrm + 1e-6))
    opt.step()
    ref.step()
    torch.testing.assert_close(p, p_ref, rtol=1e-4, atol=1e-6)
