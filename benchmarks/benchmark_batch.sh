#!/usr/bin/env bash
# Grid sweep over the shuffle benchmark (parity with the reference's
# benchmark_batch.sh:9-38 sweep: files x trainers x reducers-per-trainer,
# fixed rows/batch/trials/epochs). The reference swept 4e8 rows over a 4-node
# AWS cluster; scale NUM_ROWS to the box via env (defaults sized for a
# single MI355X node).
set -euo pipefail
cd "$(dirname "$0")"

NUM_ROWS=${NUM_ROWS:-400000000}
NUM_ROW_GROUPS_PER_FILE=${NUM_ROW_GROUPS_PER_FILE:-5}
BATCH_SIZE=${BATCH_SIZE:-250000}
NUM_TRIALS=${NUM_TRIALS:-2}
NUM_EPOCHS=${NUM_EPOCHS:-10}
MAX_CONCURRENT_EPOCHS=${MAX_CONCURRENT_EPOCHS:-2}
STATS_DIR=${STATS_DIR:-./results}

for num_files in 100 50 25; do
  for num_trainers in 16 8 4; do
    for reducers_per_trainer in 4 3 2; do
      num_reducers=$((num_trainers * reducers_per_trainer))
      echo "=== files=${num_files} trainers=${num_trainers}" \
           "reducers=${num_reducers} ==="
      python benchmark.py \
        --num-rows "${NUM_ROWS}" \
        --num-files "${num_files}" \
        --num-row-groups-per-file "${NUM_ROW_GROUPS_PER_FILE}" \
        --batch-size "${BATCH_SIZE}" \
        --num-trials "${NUM_TRIALS}" \
        --num-epochs "${NUM_EPOCHS}" \
        --max-concurrent-epochs "${MAX_CONCURRENT_EPOCHS}" \
        --num-trainers "${num_trainers}" \
        --num-reducers "${num_reducers}" \
        --stats-dir "${STATS_DIR}" \
        --use-old-data
    done
  done
done
