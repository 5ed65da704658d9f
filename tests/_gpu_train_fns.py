"""GPU training functions for engine tests (top-level for spawn pickling)."""
import torch


def gpu_trial_fn(hparams, reporter):
    """Tiny model trained with the fused HIP optimizer inside a pool worker
    (each worker sees exactly one GPU via HIP_VISIBLE_DEVICES)."""
    from maggy_amd.models import MLP
    from maggy_amd.ops import FusedAdam

    assert torch.cuda.device_count() == 1, "worker must be pinned to 1 GPU"
    device = torch.device("cuda:0")
    model = MLP(in_features=64, hidden=32, num_classes=4).to(device)
    opt = FusedAdam(model.parameters(), lr=hparams["lr"])
    x = torch.randn(32, 64, device=device)
    y = torch.randint(0, 4, (32,), device=device)
    last = None
    for step in range(6):
        opt.zero_grad(set_to_none=True)
        loss = torch.nn.functional.cross_entropy(model(x), y)
        loss.backward()
        opt.step()
        last = float(loss)
        # broadcast a CUDA tensor: exercises the HIP metric reduction
        per_sample = torch.nn.functional.cross_entropy(
            model(x), y, reduction="none").detach()
        reporter.broadcast(per_sample, step)
    return last
