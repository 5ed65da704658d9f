"""Ablation-study building blocks for tests (top-level, picklable)."""
import torch

from maggy_amd.models import SmallTransformer


def model_gen():
    torch.manual_seed(0)
    return SmallTransformer(vocab_size=50, dim=16, n_heads=2, n_layers=3,
                            num_classes=2, max_seq_len=8)


def dataset_gen(ablated_feature="None"):
    torch.manual_seed(1)
    x = torch.randint(0, 50, (64, 8))
    y = (x.sum(1) % 2).long()
    if ablated_feature == "tail":
        x = x[:, :4]
    return x, y


def ablation_train_fn(model, dataset, hparams, reporter):
    x, y = dataset
    opt = torch.optim.SGD(model.parameters(), lr=0.05)
    loss_fn = torch.nn.CrossEntropyLoss()
    last = None
    for step in range(5):
        opt.zero_grad()
        loss = loss_fn(model(x), y)
        loss.backward()
        opt.step()
        last = float(loss)
        reporter.broadcast(last, step)
    # count parameters so the test can verify layers were really dropped
    n_params = sum(p.numel() for p in model.parameters())
    return {"Metric": last, "n_params": n_params}
