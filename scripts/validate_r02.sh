#!/bin/bash
# Round-2 defaults validation (TILE 8192/1024 + rank-atomic + packed) on
# every config, plus t4k regression A/B and the 1B-pair lookback-gate A/B.
set -u
cd "$(dirname "$0")/.."
OUT=gpurun_out
mkdir -p $OUT

echo "--- full GPU test suite (new defaults) ---"
python -m pytest tests -m gpu -x -q > $OUT/pytest_gpu_r02b.log 2>&1
echo "pytest rc=$?"; tail -2 $OUT/pytest_gpu_r02b.log

run() { # name, extra bench args..., env via VARS=
  local name=$1; shift
  python bench.py "$@" --skip-cpu-baseline > $OUT/v_${name}.json 2> $OUT/v_${name}.log
  echo "== $name rc=$?: $(grep -o '"value": [0-9.]*' $OUT/v_${name}.json | head -1)" \
       "$(grep -o '"ms_per_launch": [0-9.]*' $OUT/v_${name}.json | head -1)" \
       "$(grep -o '"frac": [0-9.]*' $OUT/v_${name}.json | head -1)"
}

echo "--- all configs, new defaults ---"
run reduce --steps 10 --warmup 3
run sort   --op sort --steps 3 --warmup 1
run join   --op join --steps 3 --warmup 1 --n-pairs 250000000
run zipf   --op zipf --steps 3 --warmup 1
run reduce1b --steps 5 --warmup 2 --n-pairs 1000000000

echo "--- t4k regression A/B (sort + join) ---"
DPRK_SO=$PWD/dpark_amd/libdprk_t4k.so python bench.py --op sort --steps 3 \
  --warmup 1 --skip-cpu-baseline > $OUT/v_sort_t4k.json 2> $OUT/v_sort_t4k.log
echo "== sort_t4k rc=$?: $(grep -o '"value": [0-9.]*' $OUT/v_sort_t4k.json | head -1)"
DPRK_SO=$PWD/dpark_amd/libdprk_t4k.so python bench.py --op join --steps 3 \
  --warmup 1 --n-pairs 250000000 --skip-cpu-baseline > $OUT/v_join_t4k.json 2> $OUT/v_join_t4k.log
echo "== join_t4k rc=$?: $(grep -o '"value": [0-9.]*' $OUT/v_join_t4k.json | head -1)"
DPRK_SO=$PWD/dpark_amd/libdprk_t4k.so python bench.py --steps 5 \
  --warmup 2 --skip-cpu-baseline > $OUT/v_reduce_t4k.json 2> $OUT/v_reduce_t4k.log
echo "== reduce_t4k rc=$?: $(grep -o '"value": [0-9.]*' $OUT/v_reduce_t4k.json | head -1)"

echo "--- 1B uniform reduce: lookback-gate A/B (granules=250MB at t8k) ---"
DPRK_LB_GATE_MB=300 python bench.py --steps 5 --warmup 2 --n-pairs 1000000000 \
  --skip-cpu-baseline > $OUT/v_reduce1b_gate300.json 2> $OUT/v_reduce1b_gate300.log
echo "== reduce1b_gate300 rc=$?: $(grep -o '"value": [0-9.]*' $OUT/v_reduce1b_gate300.json | head -1)" \
     "$(grep -o '"frac": [0-9.]*' $OUT/v_reduce1b_gate300.json | head -1)"

echo "--- 1B sort: gate A/B (keys-only granules=1B/8192*2KB=250MB) ---"
DPRK_LB_GATE_MB=300 python bench.py --op sort --steps 3 --warmup 1 \
  --skip-cpu-baseline > $OUT/v_sort_gate300.json 2> $OUT/v_sort_gate300.log
echo "== sort_gate300 rc=$?: $(grep -o '"value": [0-9.]*' $OUT/v_sort_gate300.json | head -1)"
echo DONE
