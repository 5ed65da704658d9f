#!/bin/bash
# Profile the ResNet bench; only the small summary comes back.
set -x
cd /root/repo
mkdir -p gpurun_out
rm -rf /tmp/prof
export TMPDIR=/tmp
cd /tmp
MIOPEN_FIND_MODE=FAST timeout 600 rocprofv3 --kernel-trace --stats -d /tmp/prof -o bench -- python /root/repo/bench.py --gpus 1 --steps 5 --warmup 3 > /tmp/rocprof.log 2>&1
echo "rocprof exit: $?"
grep -o '{"metric.*}' /tmp/rocprof.log > $GRAFT_REPO_ROOT/gpurun_out/bench_prof.json
PYTHONPATH=/root/repo python /root/repo/scripts/prof_summarize.py /tmp/prof/bench_results.db $GRAFT_REPO_ROOT/gpurun_out/prof_summary.txt 210 > /dev/null
cat $GRAFT_REPO_ROOT/gpurun_out/prof_summary.txt
