#!/bin/bash
set -x
cd /root/repo
mkdir -p gpurun_out
python -m pytest tests/test_llama_ops_gpu.py -x -q > gpurun_out/pytest8.log 2>&1
echo "pytest exit: $?" >> gpurun_out/pytest8.log
timeout 900 python bench.py --gpus 1 --model llama1b --steps 10 --warmup 3 > gpurun_out/l1b_plain.json 2>/dev/null
PYTORCH_TUNABLEOP_ENABLED=1 PYTORCH_TUNABLEOP_TUNING=1 PYTORCH_TUNABLEOP_FILENAME=/tmp/tuned.csv \
  timeout 1200 python bench.py --gpus 1 --model llama1b --steps 10 --warmup 12 > gpurun_out/l1b_tuned.json 2> gpurun_out/l1b_tune.err
echo "tuned exit: $?" >> gpurun_out/l1b_tune.err
cp /tmp/tuned.csv gpurun_out/tunableop_1b.csv 2>/dev/null
cat gpurun_out/l1b_plain.json gpurun_out/l1b_tuned.json
tail -4 gpurun_out/pytest8.log
