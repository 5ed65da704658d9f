"""Trial functions for bench.py --mode asha (top-level for spawn pickling)."""
import torch


def resnet_trial_fn(hparams, reporter):
    """Short ResNet-50 synthetic training trial; budget = ASHA rung epochs,
    each epoch = 8 optimizer steps at batch 128."""
    from maggy_amd.models import resnet50
    from maggy_amd.ops import FusedSGD

    device = torch.device("cuda:0")
    # immediate-mode conv selection (MIOPEN_FIND_MODE=FAST heuristics):
    # benchmark=True would run a ~2 min exhaustive MIOpen find inside the
    # first trial, which belongs to the measured trials/hr
    torch.backends.cudnn.benchmark = False
    model = resnet50().to(device, memory_format=torch.channels_last)
    opt = FusedSGD(model.parameters(), lr=hparams["lr"],
                   momentum=hparams.get("momentum", 0.9))
    loss_fn = torch.nn.CrossEntropyLoss()
    batch = 128
    x = torch.randn(batch, 3, 224, 224, device=device).to(
        memory_format=torch.channels_last)
    y = torch.randint(0, 1000, (batch,), device=device)
    budget = int(hparams.get("budget", 1))
    last = None
    for epoch in range(budget):
        for _ in range(8):
            opt.zero_grad(set_to_none=True)
            with torch.autocast("cuda", dtype=torch.bfloat16):
                loss = loss_fn(model(x), y)
            loss.backward()
            opt.step()
        last = float(loss.detach())
        reporter.broadcast(last, epoch)
    return last
