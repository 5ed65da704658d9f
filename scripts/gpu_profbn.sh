#!/bin/bash
set -x
cd /root/repo
mkdir -p gpurun_out
export TMPDIR=/tmp
cd /tmp
rm -rf /tmp/profbn
timeout 600 rocprofv3 --kernel-trace --stats -d /tmp/profbn -o bnb -- env PYTHONPATH=/root/repo python /root/repo/scripts/bnbench.py > /tmp/bnb.log 2>&1
echo "exit $?"
PYTHONPATH=/root/repo python /root/repo/scripts/prof_summarize.py /tmp/profbn/bnb_results.db $GRAFT_REPO_ROOT/gpurun_out/bnprof.txt 999999 > /dev/null
tail -6 /tmp/bnb.log > $GRAFT_REPO_ROOT/gpurun_out/bnb.log
cat $GRAFT_REPO_ROOT/gpurun_out/bnprof.txt | head -20
