#!/bin/bash
# Round-2 committed evidence run (profiles/r02): kernel-trace --stats of
# the headline bench (per-kernel averages must corroborate the bench's
# roofline.ms_per_launch), PMC FETCH_SIZE / WRITE_SIZE passes (separate
# — TCC has 4 slots; FETCH costs 3, WRITE 2), and SQ stall/LDS counters
# for the (new default) scatter.
set -u
cd "$(dirname "$0")/.."
OUT=gpurun_out
mkdir -p $OUT
export TMPDIR=/tmp
ROOT=$PWD

echo "--- kernel trace + stats (config 2 bench) ---"
(cd /tmp && rocprofv3 --kernel-trace --stats -d $ROOT/$OUT/trace_r02 \
    -o trace -- python $ROOT/bench.py --steps 5 --warmup 2 \
    --skip-cpu-baseline > $ROOT/$OUT/trace_r02.log 2>&1)
echo "trace rc=$?"
grep -o '"value": [0-9.]*\|"ms_per_launch": [0-9.]*\|"frac": [0-9.]*' \
    $ROOT/$OUT/trace_r02.log | head -3

echo "--- PMC FETCH_SIZE (config 2) ---"
(cd /tmp && rocprofv3 --pmc FETCH_SIZE -d $ROOT/$OUT/pmc_fetch_r02 \
    -o fetch -- python $ROOT/bench.py --steps 3 --warmup 1 \
    --skip-cpu-baseline > $ROOT/$OUT/pmc_fetch_r02.log 2>&1)
echo "fetch rc=$?"
echo "--- PMC WRITE_SIZE (config 2) ---"
(cd /tmp && rocprofv3 --pmc WRITE_SIZE -d $ROOT/$OUT/pmc_write_r02 \
    -o write -- python $ROOT/bench.py --steps 3 --warmup 1 \
    --skip-cpu-baseline > $ROOT/$OUT/pmc_write_r02.log 2>&1)
echo "write rc=$?"
echo "--- PMC SQ stall + LDS (config 2, new default kernel) ---"
(cd /tmp && rocprofv3 --pmc SQ_WAVE_CYCLES SQ_WAIT_ANY SQ_WAIT_INST_ANY \
    SQ_ACTIVE_INST_ANY SQ_LDS_BANK_CONFLICT SQ_LDS_IDX_ACTIVE \
    -d $ROOT/$OUT/pmc_sq_r02 -o sq -- python $ROOT/bench.py --steps 3 \
    --warmup 1 --skip-cpu-baseline > $ROOT/$OUT/pmc_sq_r02.log 2>&1)
echo "sq rc=$?"
ls -la $OUT/trace_r02 $OUT/pmc_fetch_r02 $OUT/pmc_write_r02 $OUT/pmc_sq_r02 2>/dev/null | head -20
echo DONE
