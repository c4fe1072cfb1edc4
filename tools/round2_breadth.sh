#!/usr/bin/env bash
# Breadth batch: loader-only numbers, sweep grid, fp32 datapoint.
set -x
cd "$(dirname "$0")/.."
REPO="$PWD"
mkdir -p gpurun_out

# 1. Loader-only (reference trial formula), cached + uncached ingest.
for mode in auto none; do
  timeout 420 python benchmarks/benchmark.py --data-spec float100 \
    --num-rows 12500000 --num-files 4 --num-row-groups-per-file 8 \
    --batch-size 250000 --num-trials 2 --num-epochs 6 \
    --max-concurrent-epochs 2 --num-trainers 1 --num-reducers 4 \
    --source-cache $mode --use-old-data \
    --stats-dir gpurun_out/loader_stats_$mode 2>&1 | tail -6
done

# 2. Reference sweep grid (files x trainers x reducers/trainer), scaled.
NUM_ROWS=20000000 NUM_EPOCHS=4 NUM_TRIALS=1 \
  STATS_DIR="$REPO/gpurun_out/sweep_results" \
  timeout 900 bash benchmarks/benchmark_batch.sh > gpurun_out/sweep.log 2>&1
tail -5 gpurun_out/sweep.log
ls gpurun_out/sweep_results/ || true

# 3. fp32-feature datapoint at driver flags.
timeout 420 python bench.py --gpus 1 --steps 20 --warmup 5 --dtype fp32 \
  2> gpurun_out/bench_fp32.err | tee gpurun_out/bench_fp32.json
echo DONE
