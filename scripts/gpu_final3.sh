#!/bin/bash
set -x
cd /root/repo
mkdir -p gpurun_out
python -m pytest tests/ -x -q -m gpu > gpurun_out/pytest_f3.log 2>&1
echo "pytest exit: $?" >> gpurun_out/pytest_f3.log
python -c 'import __graft_entry__; __graft_entry__.smoke()' >> gpurun_out/pytest_f3.log 2>&1
echo "smoke exit: $?" >> gpurun_out/pytest_f3.log
timeout 900 python bench.py > gpurun_out/bench_default.json 2> gpurun_out/bench_default.err
echo "bench exit: $?" >> gpurun_out/bench_default.err
cat gpurun_out/bench_default.json
tail -4 gpurun_out/pytest_f3.log
