#!/bin/bash
# Round-2 scatter A/B + stall profile (run on the GPU box via gpurun).
# Variants: packed-u32 vs int64 keys (DPRK_NO_PACK), ballot vs
# ds_add_rtn ranking (DPRK_RANK_ATOMIC), TILE 4096/512 vs 8192/1024
# (DPRK_SO=libdprk_t8k.so).  Short same-box runs; winners become
# defaults, losers go in the committed A/B log.
set -u
cd "$(dirname "$0")/.."
OUT=gpurun_out
mkdir -p $OUT

bench2() { # name, env...
  local name=$1; shift
  env "$@" python bench.py --steps 5 --warmup 2 --skip-cpu-baseline \
      > $OUT/ab_${name}.json 2> $OUT/ab_${name}.log
  echo "== $name rc=$?: $(grep -o '"value": [0-9.]*' $OUT/ab_${name}.json | head -1)" \
       "$(grep -o '"ms_per_launch": [0-9.]*' $OUT/ab_${name}.json | head -1)"
}

echo "--- config 2 A/B matrix (100M reduce) ---"
bench2 base               DPRK_DUMMY=0
bench2 atomic             DPRK_RANK_ATOMIC=1
bench2 nopack             DPRK_NO_PACK=1
bench2 nopack_atomic      DPRK_NO_PACK=1 DPRK_RANK_ATOMIC=1
bench2 t8k                DPRK_SO=dpark_amd/libdprk_t8k.so
bench2 t8k_atomic         DPRK_SO=dpark_amd/libdprk_t8k.so DPRK_RANK_ATOMIC=1
bench2 t8k_nopack         DPRK_SO=dpark_amd/libdprk_t8k.so DPRK_NO_PACK=1

echo "--- config 3 sort (1B keys, 3-kernel path) ---"
env DPRK_DUMMY=0 python bench.py --op sort --steps 2 --warmup 1 \
    --skip-cpu-baseline > $OUT/ab_sort_base.json 2> $OUT/ab_sort_base.log
echo "== sort_base rc=$?: $(grep -o '"value": [0-9.]*' $OUT/ab_sort_base.json)"
env DPRK_RANK_ATOMIC=1 python bench.py --op sort --steps 2 --warmup 1 \
    --skip-cpu-baseline > $OUT/ab_sort_atomic.json 2> $OUT/ab_sort_atomic.log
echo "== sort_atomic rc=$?: $(grep -o '"value": [0-9.]*' $OUT/ab_sort_atomic.json)"

echo "--- rank-atomic parity guard ---"
env DPRK_RANK_ATOMIC=1 python -m pytest tests/test_gpu.py -x -q \
    -k "golden or determinism or parity or soak or packed" \
    > $OUT/pytest_rank_atomic.log 2>&1
echo "rank_atomic pytest rc=$?"; tail -2 $OUT/pytest_rank_atomic.log
env DPRK_SO=$PWD/dpark_amd/libdprk_t8k.so python -m pytest tests/test_gpu.py \
    -x -q -k "golden or determinism" > $OUT/pytest_t8k.log 2>&1
echo "t8k pytest rc=$?"; tail -2 $OUT/pytest_t8k.log

echo "--- wordcount leg (string path) ---"
python bench.py --op wordcount --steps 3 --warmup 1 \
    > $OUT/bench_wordcount.json 2> $OUT/bench_wordcount.log
echo "wordcount rc=$?"; cat $OUT/bench_wordcount.json

echo "--- PMC stall profile of the scatter (config 2 short) ---"
export TMPDIR=/tmp
ROOT=$PWD
(cd /tmp && rocprofv3 --pmc SQ_WAVE_CYCLES SQ_WAIT_ANY SQ_WAIT_INST_ANY \
    SQ_ACTIVE_INST_ANY -d $ROOT/$OUT/pmc_stall -o stall -- \
    python $ROOT/bench.py --steps 3 --warmup 1 --skip-cpu-baseline \
    > $ROOT/$OUT/pmc_stall.log 2>&1)
echo "pmc_stall rc=$?"
(cd /tmp && rocprofv3 --pmc SQ_LDS_BANK_CONFLICT SQ_LDS_IDX_ACTIVE \
    SQ_WAVE_CYCLES -d $ROOT/$OUT/pmc_lds -o lds -- \
    python $ROOT/bench.py --steps 3 --warmup 1 --skip-cpu-baseline \
    > $ROOT/$OUT/pmc_lds.log 2>&1)
echo "pmc_lds rc=$?"
# compact the counter CSVs (keep only scatter/reduce/gen rows)
for d in pmc_stall pmc_lds; do
  f=$(find $OUT/$d -name "*counter_collection.csv" | head -1)
  [ -n "$f" ] && head -1 "$f" > $OUT/${d}_summary.csv && \
    grep -E "scatter|reduce_tiles|gen_hist" "$f" >> $OUT/${d}_summary.csv && \
    wc -l $OUT/${d}_summary.csv
done

echo "--- zipf config 5 (1B, packed path validation) ---"
python bench.py --op zipf --steps 3 --warmup 1 --skip-cpu-baseline \
    > $OUT/bench_zipf.json 2> $OUT/bench_zipf.log
echo "zipf rc=$?"; cat $OUT/bench_zipf.json
echo DONE
