"""LOCO ablation-study example on a small Transformer.

One trial per ablated component (plus the base run): layer groups are
dropped from the nn.Module, features from the dataset — the reference's
Keras-JSON surgery (/root/reference/maggy/ablation/ablator/loco.py:99-136)
replaced by nn.Module child dropping.

    python examples/ablation_transformer.py
"""
import argparse
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch  # noqa: E402

from maggy_amd import experiment  # noqa: E402
from maggy_amd.ablation import AblationStudy  # noqa: E402
from maggy_amd.config import AblationConfig  # noqa: E402


def model_generator(ablated_layer="None"):
    """Rebuild the model minus the ablated layer (group)."""
    from maggy_amd.models import SmallTransformer
    from maggy_amd.ablation import drop_layers

    torch.manual_seed(0)
    model = SmallTransformer(vocab_size=100, dim=32, n_heads=4, n_layers=4,
                             num_classes=2, max_seq_len=16)
    if ablated_layer not in (None, "None"):
        model = drop_layers(model, ablated_layer)
    return model


def dataset_generator(ablated_feature="None"):
    torch.manual_seed(1)
    x = torch.randint(0, 100, (256, 16))
    y = (x.sum(1) % 2).long()
    if ablated_feature == "suffix_tokens":
        x = x[:, :8]  # drop the input's tail half
    return x, y


def train_fn(model, dataset, hparams, reporter):
    x, y = dataset
    opt = torch.optim.Adam(model.parameters(), lr=1e-3)
    loss_fn = torch.nn.CrossEntropyLoss()
    for step in range(8):
        opt.zero_grad()
        loss = loss_fn(model(x), y)
        loss.backward()
        opt.step()
        reporter.broadcast(float(loss), step)
    return {"Metric": float(loss),
            "n_params": sum(p.numel() for p in model.parameters())}


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--workers", type=int, default=2)
    args = ap.parse_args()
    os.environ.setdefault("MAGGY_LOG_DIR", "./maggy_logs")

    study = AblationStudy(model_generator=model_generator,
                          dataset_generator=dataset_generator)
    study.features.include("suffix_tokens")
    study.model.layers.include("blocks.0", "blocks.3")
    study.model.layers.include_groups(["blocks.1", "blocks.2"])
    config = AblationConfig(ablation_study=study, ablator="loco",
                            direction="min", num_workers=args.workers,
                            name="example-ablation")
    result = experiment.lagom(train_fn, config)
    # base trial + 1 feature + 2 layers + 1 group = 5 trials
    print("trials:", result["num_trials"], "best:", result["best_config"])


if __name__ == "__main__":
    main()
