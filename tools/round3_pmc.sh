#!/usr/bin/env bash
# Round-3 stall-class diagnostic (ROADMAP item 3 "next diagnostic"):
# SQ wait/busy breakdown of the chain kernels, baseline vs the pinned
# variants, one run per counter group (rocprofv3 --pmc must not be
# combined with sys/runtime/hip/hsa trace domains on this pool).
# Usage: /usr/local/graft/bin/gpurun --timeout 1500 -- 'bash tools/round3_pmc.sh'
set -x
cd "$(dirname "$0")/.."
REPO="$PWD"
mkdir -p gpurun_out
export TMPDIR=/tmp
cd /tmp

run_pmc() {  # run_pmc <tag> <env...> -- <counters...>
  local tag="$1"; shift
  local envs=()
  while [ "$1" != "--" ]; do envs+=("$1"); shift; done
  shift
  RSDL_PROF_ITERS=5 env "${envs[@]}" timeout 300 rocprofv3 --pmc "$@" \
    --kernel-trace --stats -d "$REPO/gpurun_out/pmc_$tag" -- \
    python "$REPO/tools/profile_fused_step.py" \
    > "$REPO/gpurun_out/pmc_$tag.log" 2>&1 \
    || tail -5 "$REPO/gpurun_out/pmc_$tag.log"
}

# Stall classes: where do the chain waves wait? (SQ_WAIT_ANY was 75-80%
# of SQ_BUSY_CYCLES in round 2; split it by class.)
run_pmc wait_base -- SQ_WAIT_ANY SQ_WAIT_INST_LDS SQ_BUSY_CYCLES SQ_WAVES
run_pmc wait2_base -- SQ_INST_CYCLES_VMEM SQ_INSTS_VALU SQ_INSTS_MFMA SQ_ACTIVE_INST_VALU

# Same two groups with the pinned-schedule + pi16 variants on.
run_pmc wait_sched RSDL_WGRAD_SCHED=1 RSDL_PI16=1 -- \
  SQ_WAIT_ANY SQ_WAIT_INST_LDS SQ_BUSY_CYCLES SQ_WAVES
run_pmc wait2_sched RSDL_WGRAD_SCHED=1 RSDL_PI16=1 -- \
  SQ_INST_CYCLES_VMEM SQ_INSTS_VALU SQ_INSTS_MFMA SQ_ACTIVE_INST_VALU

echo DONE
