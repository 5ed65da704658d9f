#!/bin/bash
set -x
cd /root/repo
mkdir -p gpurun_out
{ time python -m pytest tests -m gpu -q ; } > gpurun_out/pytest_final.log 2>&1
echo "pytest exit: $?" >> gpurun_out/pytest_final.log
python -c 'import __graft_entry__; __graft_entry__.smoke()' > gpurun_out/smoke_final.log 2>&1
echo "smoke exit: $?" >> gpurun_out/smoke_final.log
timeout 600 python bench.py --gpus 1 --steps 20 --warmup 10 > gpurun_out/final_r50.json 2>/dev/null
timeout 600 python bench.py --gpus 1 --steps 20 --warmup 10 --graphs > gpurun_out/final_r50_graphs.json 2> gpurun_out/graphs.err
echo "graphs exit: $?" >> gpurun_out/graphs.err
timeout 900 python bench.py --mode asha --gpus 1 > gpurun_out/final_asha.json 2> gpurun_out/asha.err
echo "asha exit: $?" >> gpurun_out/asha.err
cat gpurun_out/final_r50.json gpurun_out/final_r50_graphs.json gpurun_out/final_asha.json
tail -4 gpurun_out/pytest_final.log gpurun_out/smoke_final.log
tail -3 gpurun_out/graphs.err gpurun_out/asha.err
