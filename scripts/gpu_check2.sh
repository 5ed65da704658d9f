#!/bin/bash
# Validate fused BN kernels + measure the ResNet bench delta.
set -x
cd /root/repo
mkdir -p gpurun_out
{ time python -m pytest tests/test_fused_bn_gpu.py tests/test_ops_gpu.py -x -q ; } > gpurun_out/pytest_bn.log 2>&1
echo "pytest exit: $?" >> gpurun_out/pytest_bn.log
MIOPEN_FIND_MODE=FAST timeout 600 python bench.py --gpus 1 --steps 20 --warmup 10 > gpurun_out/bench_bn.json 2> gpurun_out/bench_bn.err
echo "bench exit: $?" >> gpurun_out/bench_bn.err
export TMPDIR=/tmp
cd /tmp
MIOPEN_FIND_MODE=FAST timeout 600 rocprofv3 --kernel-trace --stats -d $GRAFT_REPO_ROOT/gpurun_out/prof2 -o bench2 -- python /root/repo/bench.py --gpus 1 --steps 5 --warmup 3 > $GRAFT_REPO_ROOT/gpurun_out/rocprof2.log 2>&1
echo "rocprof exit: $?" >> $GRAFT_REPO_ROOT/gpurun_out/rocprof2.log
cat $GRAFT_REPO_ROOT/gpurun_out/bench_bn.json
tail -15 $GRAFT_REPO_ROOT/gpurun_out/pytest_bn.log
