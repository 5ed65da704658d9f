#!/bin/bash
set -x
cd /root/repo
mkdir -p gpurun_out
python -m pytest tests -m gpu -q > gpurun_out/pytest_f2.log 2>&1
echo "pytest exit: $?" >> gpurun_out/pytest_f2.log
timeout 600 python bench.py --gpus 1 --steps 20 --warmup 10 > gpurun_out/f2_r50.json 2>/dev/null
timeout 900 python bench.py --gpus 1 --model llama8b --steps 5 --warmup 2 > gpurun_out/f2_l8b.json 2>/dev/null
timeout 600 python bench.py --mode asha --gpus 1 > gpurun_out/f2_asha.json 2>/dev/null || true
grep -h '"metric"' gpurun_out/f2_r50.json gpurun_out/f2_l8b.json gpurun_out/f2_asha.json
tail -3 gpurun_out/pytest_f2.log
