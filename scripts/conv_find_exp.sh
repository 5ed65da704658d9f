#!/bin/bash
# ResNet conv-algorithm experiment (round-2 VERDICT #7): does exhaustive
# MIOpen find beat the FAST-mode heuristics at b512?  Run on a GPU box:
#
#   bash scripts/conv_find_exp.sh
#
# Arms (same bench step, steady-state measured after warmup):
#   A. MIOPEN_FIND_MODE=FAST (current default)
#   B. MIOPEN_FIND_MODE=NORMAL + torch.backends.cudnn.benchmark=True
#      (exhaustive per-shape find, cached in the user find-db)
#   C. re-run of B with the warm find-db (what shipping the db would give)
set -x
mkdir -p gpurun_out/convdb
export MIOPEN_USER_DB_PATH=$PWD/gpurun_out/convdb
export MIOPEN_CUSTOM_CACHE_DIR=$PWD/gpurun_out/convdb

MIOPEN_FIND_MODE=FAST timeout 500 python bench.py --steps 15 --warmup 5 \
    > gpurun_out/conv_fast.json 2>&1
MIOPEN_FIND_MODE=NORMAL MAGGY_CUDNN_BENCHMARK=1 timeout 900 \
    python bench.py --steps 15 --warmup 8 \
    > gpurun_out/conv_find.json 2>&1
MIOPEN_FIND_MODE=NORMAL MAGGY_CUDNN_BENCHMARK=1 timeout 500 \
    python bench.py --steps 15 --warmup 5 \
    > gpurun_out/conv_find_warm.json 2>&1
ls -la $MIOPEN_USER_DB_PATH
tail -1 gpurun_out/conv_fast.json
tail -1 gpurun_out/conv_find.json
tail -1 gpurun_out/conv_find_warm.json
