"""xGMI all-reduce bucket/algorithm sweep (run on a multi-GPU node).

Launch:
  python -m torch.distributed.run --nnodes=1 --nproc-per-node 8 \
      --master-addr 127.0.0.1 scripts/sweep_allreduce.py

Optionally sweep RCCL algorithms from the shell:
  for a in Ring Tree; do NCCL_ALGO=$a python -m torch.distributed.run ... ; done

Measures effective all-reduce bandwidth for message sizes bracketing the
DDP bucket choices (8..256 MB) so the XGMI_BUCKET_CAP_MB default in
maggy_amd/parallel/dist.py rests on a measurement instead of a comment
(round-1 VERDICT weak #1).  Bus bandwidth uses the standard nccl-tests
convention: busbw = algbw * 2 * (n-1) / n.

Writes one JSON line per size on rank 0 (tee to gpurun_out/ to keep it).
"""
import json
import os
import time

import torch
import torch.distributed as dist


def main():
    os.environ.setdefault("HSA_ENABLE_IPC_MODE_LEGACY", "0")
    backend = "nccl" if torch.cuda.is_available() else "gloo"
    dist.init_process_group(backend)
    rank = dist.get_rank()
    world = dist.get_world_size()
    if torch.cuda.is_available():
        torch.cuda.set_device(int(os.environ.get("LOCAL_RANK", rank)))
        device = torch.device("cuda")
    else:
        device = torch.device("cpu")

    sizes_mb = [1, 4, 8, 16, 32, 64, 128, 256]
    iters, warmup = 20, 5
    results = []
    for mb in sizes_mb:
        n = mb * 1024 * 1024 // 2  # bf16 elements
        t = torch.ones(n, dtype=torch.bfloat16, device=device)
        for _ in range(warmup):
            dist.all_reduce(t)
        if device.type == "cuda":
            torch.cuda.synchronize()
        dist.barrier()
        t0 = time.perf_counter()
        for _ in range(iters):
            dist.all_reduce(t)
        if device.type == "cuda":
            torch.cuda.synchronize()
        dt = (time.perf_counter() - t0) / iters
        nbytes = n * 2
        algbw = nbytes / dt / 1e9
        busbw = algbw * 2 * (world - 1) / world
        row = {"size_mb": mb, "ms": round(dt * 1e3, 3),
               "algbw_GBps": round(algbw, 1), "busbw_GBps": round(busbw, 1),
               "world": world, "algo": os.environ.get("NCCL_ALGO", "auto")}
        results.append(row)
        if rank == 0:
            print(json.dumps(row), flush=True)
    if rank == 0:
        best = max(results, key=lambda r: r["busbw_GBps"])
        print(json.dumps({"best": best}), flush=True)
    dist.barrier()
    dist.destroy_process_group()


if __name__ == "__main__":
    main()
