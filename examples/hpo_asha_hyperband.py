"""Multi-fidelity HPO example: ASHA, and RandomSearch under a Hyperband
pruner with promoted-trial checkpoint continuation.

The controller injects ``budget`` into the hparams; promoted trials get
``parent_checkpoint`` so they CONTINUE from the parent's weights instead
of restarting (the reference restarts promoted configs from scratch —
continuation is a capability on top of parity).

    python examples/hpo_asha_hyperband.py [--mode asha|hyperband]
"""
import argparse
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch  # noqa: E402

from maggy_amd import Searchspace, experiment  # noqa: E402
from maggy_amd.config import HyperparameterOptConfig  # noqa: E402
from maggy_amd.optimizer import Asha, RandomSearch  # noqa: E402


def budget_train_fn(hparams, reporter, trial_dir, parent_checkpoint):
    """Budgeted trial: run ``budget`` epochs; save/continue checkpoints."""
    from maggy_amd.utils.checkpoint import save_checkpoint

    torch.manual_seed(7)
    model = torch.nn.Sequential(
        torch.nn.Linear(16, 32), torch.nn.ReLU(), torch.nn.Linear(32, 2))
    start_epoch = 0
    if parent_checkpoint is not None:
        state = torch.load(parent_checkpoint, weights_only=False)
        model.load_state_dict(state["model"])
        start_epoch = state.get("step", 0)
    opt = torch.optim.SGD(model.parameters(), lr=hparams["lr"])
    x = torch.randn(128, 16)
    y = (x.sum(1) > 0).long()
    loss_fn = torch.nn.CrossEntropyLoss()
    budget = int(hparams.get("budget", 1))
    loss = None
    for epoch in range(start_epoch, start_epoch + budget):
        for _ in range(5):
            opt.zero_grad()
            loss = loss_fn(model(x), y)
            loss.backward()
            opt.step()
        reporter.broadcast(float(loss), epoch)
    save_checkpoint(trial_dir, model, step=start_epoch + budget)
    return float(loss)


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--mode", default="asha", choices=["asha", "hyperband"])
    ap.add_argument("--workers", type=int, default=2)
    args = ap.parse_args()
    os.environ.setdefault("MAGGY_LOG_DIR", "./maggy_logs")
    sp = Searchspace(lr=("DOUBLE", [1e-3, 0.5]))
    if args.mode == "asha":
        opt = Asha(reduction_factor=2, resource_min=1, resource_max=4)
        num_trials = 12
    else:
        opt = RandomSearch(pruner="hyperband",
                           pruner_kwargs=dict(min_budget=1, max_budget=4,
                                              eta=2, n_iterations=1))
        num_trials = 7
    config = HyperparameterOptConfig(
        num_trials=num_trials, optimizer=opt, searchspace=sp,
        direction="min", es_policy="none", num_workers=args.workers,
        name="example-" + args.mode)
    result = experiment.lagom(budget_train_fn, config)
    print("best:", result["best_config"], result["best_val"])


if __name__ == "__main__":
    main()
