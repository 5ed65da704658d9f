"""KV-cached decode throughput for the Llama family (serving datapoint).

    python scripts/decode_bench.py [--size 1b|8b] [--batch 8] [--new 64]
"""
import argparse
import os
import sys
import time

import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

from maggy_amd.models import LlamaConfig, LlamaModel  # noqa: E402


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--size", default="1b", choices=["1b", "8b"])
    ap.add_argument("--batch", type=int, default=8)
    ap.add_argument("--prompt", type=int, default=512)
    ap.add_argument("--new", type=int, default=64)
    ap.add_argument("--iters", type=int, default=3)
    ap.add_argument("--captured", action="store_true",
                    help="hipGraph-captured decode step")
    args = ap.parse_args()
    cfg = (LlamaConfig.llama3_8b() if args.size == "8b"
           else LlamaConfig.small_1b())
    with torch.device("cuda"):
        model = LlamaModel(cfg)
    model = model.to(torch.bfloat16).eval()
    model.rope_cos = model.rope_cos.float()
    model.rope_sin = model.rope_sin.float()
    tokens = torch.randint(0, cfg.vocab_size, (args.batch, args.prompt),
                           device="cuda")
    gen = (lambda t, n: model.generate_captured(t, n)) if args.captured \
        else (lambda t, n: model.generate(t, n))
    # warmup
    gen(tokens, 8)
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(args.iters):
        gen(tokens, args.new)
    torch.cuda.synchronize()
    dt = (time.perf_counter() - t0) / args.iters
    new_tok = args.batch * args.new
    print("llama-%s decode%s: batch %d, prompt %d, %d new tokens: "
          "%.3f s -> %.0f tokens/sec decode (%.2f ms/token/batch)"
          % (args.size, " (captured)" if args.captured else "", args.batch, args.prompt, args.new, dt,
             new_tok / dt, dt / args.new * 1e3))


if __name__ == "__main__":
    main()
