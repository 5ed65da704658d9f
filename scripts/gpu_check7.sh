#!/bin/bash
set -x
cd /root/repo
mkdir -p gpurun_out
python -m pytest tests/test_llama_ops_gpu.py tests/test_fused_bn_gpu.py tests/test_ops_gpu.py -x -q > gpurun_out/pytest7.log 2>&1
echo "pytest exit: $?" >> gpurun_out/pytest7.log
timeout 900 python bench.py --gpus 1 --model llama1b --steps 10 --warmup 3 > gpurun_out/bench_l1b_7.json 2> gpurun_out/l1b7.err
timeout 900 python bench.py --gpus 1 --model llama8b --steps 5 --warmup 2 > gpurun_out/bench_l8b_7.json 2> gpurun_out/l8b7.err
MIOPEN_FIND_MODE=FAST timeout 600 python bench.py --gpus 1 --steps 20 --warmup 10 > gpurun_out/bench_r50_7.json 2>> gpurun_out/l8b7.err
cat gpurun_out/bench_l1b_7.json gpurun_out/bench_l8b_7.json gpurun_out/bench_r50_7.json
tail -4 gpurun_out/pytest7.log
