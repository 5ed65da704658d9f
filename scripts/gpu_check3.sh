#!/bin/bash
set -x
cd /root/repo
mkdir -p gpurun_out
python -m pytest tests/test_fused_bn_gpu.py -x -q > gpurun_out/pytest_bn3.log 2>&1
echo "pytest exit: $?" >> gpurun_out/pytest_bn3.log
python scripts/bnbench.py > gpurun_out/bnbench.log 2>&1
MIOPEN_FIND_MODE=FAST timeout 600 python bench.py --gpus 1 --steps 20 --warmup 10 > gpurun_out/bench_bn3.json 2> gpurun_out/bench_bn3.err
cat gpurun_out/bench_bn3.json
cat gpurun_out/bnbench.log
tail -3 gpurun_out/pytest_bn3.log
