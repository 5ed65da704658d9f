#!/bin/bash
set -x
cd /root/repo
mkdir -p gpurun_out
PYTHONPATH=/root/repo python scripts/bnbench.py > gpurun_out/bnbench.log 2>&1
export TMPDIR=/tmp
cd /tmp
MIOPEN_FIND_MODE=FAST timeout 600 rocprofv3 --kernel-trace --stats -d $GRAFT_REPO_ROOT/gpurun_out/prof3 -o bench3 -- python /root/repo/bench.py --gpus 1 --steps 5 --warmup 3 > $GRAFT_REPO_ROOT/gpurun_out/rocprof3.log 2>&1
echo "rocprof exit: $?"
cat $GRAFT_REPO_ROOT/gpurun_out/bnbench.log
