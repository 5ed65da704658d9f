"""Data-parallel training example: TorchDistributedConfig over the
process-per-GPU RCCL engine (gloo on CPU, same code path).

The module is passed as a CLASS (the reference contract,
/root/reference/maggy/config/torch_distributed.py:46-47): each rank
instantiates it locally; the wrapper device-places and DDP-wraps it and
the DataLoader is patched with a DistributedSampler automatically.

    python examples/dist_llama_ddp.py --workers 2            # CPU/gloo
    python examples/dist_llama_ddp.py --workers 8 --full     # 8x MI355X
    python examples/dist_llama_ddp.py --zero 2               # grad sharding
"""
import argparse
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch  # noqa: E402

from maggy_amd import experiment  # noqa: E402
from maggy_amd.config import TorchDistributedConfig  # noqa: E402


class ExampleLlama(torch.nn.Module):
    def __init__(self, full=False):
        super().__init__()
        from maggy_amd.models.llama import LlamaConfig, LlamaModel

        cfg = (LlamaConfig.small_1b() if full
               else LlamaConfig.tiny(vocab_size=256))
        self.inner = LlamaModel(cfg)
        self.vocab = cfg.vocab_size

    def forward(self, tokens):
        return self.inner(tokens, targets=tokens)  # next-token CE loss


def train_fn(module, hparams, reporter):
    import torch.distributed as dist

    full = bool(hparams.get("full"))
    torch.manual_seed(0)  # identical init on every rank
    model = module(full=full)  # device-placed + DDP-wrapped by the engine
    dev = next(model.parameters()).device
    if full:
        model = model.to(torch.bfloat16)
    opt = torch.optim.AdamW(model.parameters(), lr=hparams["lr"])
    rank = dist.get_rank()
    torch.manual_seed(100 + rank)  # per-rank data shard
    vocab = model.module.vocab if hasattr(model, "module") else model.vocab
    tokens = torch.randint(0, vocab, (2, 256 if full else 16), device=dev)
    loss = None
    for step in range(int(hparams.get("steps", 5))):
        opt.zero_grad()
        loss = model(tokens)
        loss.backward()  # bucketed all-reduce over xGMI, overlapped
        opt.step()
        reporter.broadcast(float(loss), step)
    return {"Metric": float(loss), "rank": rank}


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--workers", type=int, default=2)
    ap.add_argument("--full", action="store_true")
    ap.add_argument("--zero", type=int, default=0)
    ap.add_argument("--mixed-precision", action="store_true")
    args = ap.parse_args()
    os.environ.setdefault("MAGGY_LOG_DIR", "./maggy_logs")
    config = TorchDistributedConfig(
        module=ExampleLlama,
        hparams={"lr": 1e-4, "steps": 5, "full": args.full},
        num_gpus=args.workers, zero_lvl=args.zero,
        mixed_precision=args.mixed_precision, name="example-ddp")
    result = experiment.lagom(train_fn, config)
    print("world:", result["world_size"],
          "avg final loss:", result["final_metric_avg"])


if __name__ == "__main__":
    main()
