#!/bin/bash
set -x
cd /root/repo
mkdir -p gpurun_out
timeout 600 python bench.py --gpus 1 --steps 15 --warmup 8 --batch 512 > gpurun_out/r50_b512.json 2>/dev/null
timeout 900 python bench.py --gpus 1 --model llama8b --steps 4 --warmup 2 --batch 4 > gpurun_out/l8b_b4.json 2> gpurun_out/l8b_b4.err
echo "l8b b4 exit: $?" >> gpurun_out/l8b_b4.err
grep -h '"metric"' gpurun_out/r50_b512.json gpurun_out/l8b_b4.json 2>/dev/null
tail -2 gpurun_out/l8b_b4.err
