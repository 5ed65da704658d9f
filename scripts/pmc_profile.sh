#!/bin/bash
# PMC counter passes for the hand-written kernels (run on a GPU box).
#
#   bash scripts/pmc_profile.sh [outdir]
#
# Three separate rocprofv3 passes (never combined with trace domains —
# the pool forbids --pmc together with --sys-trace/--kernel-trace):
#   1. SQ pass: wave cycles, MFMA busy, LDS bank conflicts / LDS cycles
#   2. TCC fetch pass: FETCH_SIZE (3 slots; cannot share with WRITE_SIZE)
#   3. TCC write pass: WRITE_SIZE
# plus one --stats kernel-time pass for the same command.
#
# gfx950 note (MI355X_MICROARCH.md): FETCH_SIZE reports HALF the bytes of
# a wide coalesced streaming read — double before comparing to theory.
set -e
OUT=${1:-gpurun_out/pmc}
mkdir -p "$OUT"
cd /tmp && export TMPDIR=/tmp
cd - >/dev/null

CMD="python scripts/kernel_micro.py --iters 10"

run() {
  name=$1; shift
  rocprofv3 "$@" -d "$OUT/$name" -o "$name" -- $CMD >"$OUT/$name.log" 2>&1 \
    || echo "pass $name failed (see $OUT/$name.log)"
}

run sq   --pmc SQ_WAVE_CYCLES SQ_VALU_MFMA_BUSY_CYCLES \
         SQ_LDS_BANK_CONFLICT SQ_LDS_IDX_ACTIVE
run fetch --pmc FETCH_SIZE
run write --pmc WRITE_SIZE
run stats --stats --kernel-trace

echo "--- summaries ---"
for d in sq fetch write stats; do
  echo "== $d =="
  find "$OUT/$d" -name '*.csv' | head -4
done
