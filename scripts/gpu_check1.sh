#!/bin/bash
# First GPU validation: kernel numerics, engine e2e, smoke, 1-GPU bench, rocprof stats.
set -x
cd /root/repo
mkdir -p gpurun_out
rocm-smi --showproductname 2>/dev/null | head -5 > gpurun_out/gpu_info.txt
{ time python -m pytest tests -m gpu -x -q ; } > gpurun_out/pytest_gpu.log 2>&1
echo "pytest exit: $?" >> gpurun_out/pytest_gpu.log
python -c 'import __graft_entry__; __graft_entry__.smoke()' > gpurun_out/smoke.log 2>&1
echo "smoke exit: $?" >> gpurun_out/smoke.log
MIOPEN_FIND_MODE=FAST timeout 600 python bench.py --gpus 1 --steps 20 --warmup 10 > gpurun_out/bench_resnet1.json 2> gpurun_out/bench_resnet1.err
echo "bench exit: $?" >> gpurun_out/bench_resnet1.err
export TMPDIR=/tmp
cd /tmp
MIOPEN_FIND_MODE=FAST timeout 600 rocprofv3 --kernel-trace --stats -d $GRAFT_REPO_ROOT/gpurun_out/prof -o bench1 -- python /root/repo/bench.py --gpus 1 --steps 5 --warmup 3 > $GRAFT_REPO_ROOT/gpurun_out/rocprof.log 2>&1
echo "rocprof exit: $?" >> $GRAFT_REPO_ROOT/gpurun_out/rocprof.log
tail -5 $GRAFT_REPO_ROOT/gpurun_out/bench_resnet1.json
tail -20 $GRAFT_REPO_ROOT/gpurun_out/pytest_gpu.log
