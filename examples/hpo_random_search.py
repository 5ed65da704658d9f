"""RandomSearch HPO example: one oblivious training function, a search
space, and ``experiment.lagom`` — the reference's flagship usage
(/root/reference/README.md quickstart) on the MI355X trial pool.

    python examples/hpo_random_search.py          # tiny MLP trials, CPU
    python examples/hpo_random_search.py --full   # ResNet-50 bf16 on GPU
"""
import argparse
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch  # noqa: E402

from maggy_amd import Searchspace, experiment  # noqa: E402
from maggy_amd.config import HyperparameterOptConfig  # noqa: E402


def train_fn(hparams, reporter):
    """Oblivious training function: gets hparams, reports metrics.
    The SAME function runs on CPU or pinned to one MI355X of the pool."""
    full = bool(int(os.environ.get("EXAMPLE_FULL", "0")))
    device = "cuda" if (full and torch.cuda.is_available()) else "cpu"
    torch.manual_seed(0)
    if full:
        from maggy_amd.models import resnet50

        model = resnet50().to(device, memory_format=torch.channels_last)
        x = torch.randn(64, 3, 224, 224, device=device).to(
            memory_format=torch.channels_last)
        y = torch.randint(0, 1000, (64,), device=device)
        steps = 20
    else:
        model = torch.nn.Sequential(
            torch.nn.Linear(32, 64), torch.nn.ReLU(),
            torch.nn.Linear(64, 4)).to(device)
        x = torch.randn(256, 32, device=device)
        y = torch.randint(0, 4, (256,), device=device)
        steps = 10
    opt = torch.optim.SGD(model.parameters(), lr=hparams["lr"],
                          momentum=hparams["momentum"])
    loss_fn = torch.nn.CrossEntropyLoss()
    for step in range(steps):
        opt.zero_grad()
        if device == "cuda":
            with torch.autocast("cuda", dtype=torch.bfloat16):
                loss = loss_fn(model(x), y)
        else:
            loss = loss_fn(model(x), y)
        loss.backward()
        opt.step()
        # heartbeat: drives the live metric stream AND the median
        # early-stop rule; raises EarlyStopException when the driver
        # flags this trial
        reporter.broadcast(float(loss), step)
    return float(loss)


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--full", action="store_true")
    ap.add_argument("--trials", type=int, default=6)
    ap.add_argument("--workers", type=int, default=2)
    args = ap.parse_args()
    os.environ["EXAMPLE_FULL"] = "1" if args.full else "0"
    os.environ.setdefault("MAGGY_LOG_DIR", "./maggy_logs")

    sp = Searchspace(lr=("DOUBLE", [1e-3, 1e-1]),
                     momentum=("DOUBLE", [0.5, 0.99]))
    config = HyperparameterOptConfig(
        num_trials=args.trials, optimizer="randomsearch", searchspace=sp,
        direction="min", es_policy="median", es_interval=1, es_min=3,
        num_workers=args.workers, name="example-randomsearch")
    result = experiment.lagom(train_fn, config)
    print("best config:", result["best_config"],
          "best metric:", result["best_val"])


if __name__ == "__main__":
    main()
