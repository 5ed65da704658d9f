#!/bin/bash
set -x
cd /root/repo
mkdir -p gpurun_out
python -m pytest tests/test_fused_bn_gpu.py -x -q > gpurun_out/pytest6.log 2>&1
echo "pytest exit: $?" >> gpurun_out/pytest6.log
PYTHONPATH=/root/repo python scripts/bnbench.py > gpurun_out/bnbench6.log 2>&1
MIOPEN_FIND_MODE=FAST timeout 600 python bench.py --gpus 1 --steps 20 --warmup 10 > gpurun_out/bench6.json 2>> gpurun_out/pytest6.log
timeout 900 python bench.py --gpus 1 --model llama1b --steps 10 --warmup 3 --seq-len 4096 > gpurun_out/bench_llama1b.json 2> gpurun_out/llama1b.err
echo "llama1b exit: $?" >> gpurun_out/llama1b.err
timeout 900 python bench.py --gpus 1 --model llama8b --steps 5 --warmup 2 --seq-len 4096 > gpurun_out/bench_llama8b.json 2> gpurun_out/llama8b.err
echo "llama8b exit: $?" >> gpurun_out/llama8b.err
cat gpurun_out/bench6.json gpurun_out/bench_llama1b.json gpurun_out/bench_llama8b.json
cat gpurun_out/bnbench6.log
tail -2 gpurun_out/pytest6.log gpurun_out/llama8b.err
