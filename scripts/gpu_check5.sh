#!/bin/bash
set -x
cd /root/repo
mkdir -p gpurun_out
python -m pytest tests/test_fused_bn_gpu.py -x -q > gpurun_out/pytest_bn5.log 2>&1
echo "pytest exit: $?" >> gpurun_out/pytest_bn5.log
PYTHONPATH=/root/repo python scripts/bnbench.py > gpurun_out/bnbench5.log 2>&1
MIOPEN_FIND_MODE=FAST timeout 600 python bench.py --gpus 1 --steps 20 --warmup 10 > gpurun_out/bench5.json 2> gpurun_out/bench5.err
cat gpurun_out/bench5.json
cat gpurun_out/bnbench5.log
tail -3 gpurun_out/pytest_bn5.log
