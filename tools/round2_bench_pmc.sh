#!/usr/bin/env bash
# GPU call: numerics + driver-flag bench (fused default) + chain-kernel PMC.
set -x
cd "$(dirname "$0")/.."
REPO="$PWD"
mkdir -p gpurun_out

RSDL_EXPERIMENTAL=1 timeout 240 python -m pytest tests/test_gpu_kernels.py -m gpu -q \
    -k "fwd_chain or bwd_chain or fused_step" 2>&1 | tail -2

timeout 420 python bench.py --gpus 1 --steps 20 --warmup 5 2> gpurun_out/bench_fused.err | tee gpurun_out/bench_fused.json
tail -2 gpurun_out/bench_fused.err

export TMPDIR=/tmp
cd /tmp
RSDL_PROF_ITERS=5 timeout 300 rocprofv3 --pmc FETCH_SIZE WRITE_SIZE SQ_WAVES SQ_BUSY_CYCLES \
  --kernel-trace --stats -d "$REPO/gpurun_out/pmc_a" -- \
  python "$REPO/tools/profile_fused_step.py" > "$REPO/gpurun_out/pmc_a.log" 2>&1 || tail -5 "$REPO/gpurun_out/pmc_a.log"
RSDL_PROF_ITERS=5 timeout 300 rocprofv3 --pmc SQ_INSTS_LDS SQ_INSTS_MFMA SQ_INSTS_VALU SQ_INSTS_VMEM \
  --kernel-trace --stats -d "$REPO/gpurun_out/pmc_b" -- \
  python "$REPO/tools/profile_fused_step.py" > "$REPO/gpurun_out/pmc_b.log" 2>&1 || tail -5 "$REPO/gpurun_out/pmc_b.log"
RSDL_PROF_ITERS=5 timeout 300 rocprofv3 --pmc SQ_ACCUM_PREV_HIRES SQ_WAIT_INST_LDS SQ_INST_CYCLES_VMEM \
  --kernel-trace --stats -d "$REPO/gpurun_out/pmc_c" -- \
  python "$REPO/tools/profile_fused_step.py" > "$REPO/gpurun_out/pmc_c.log" 2>&1 || tail -5 "$REPO/gpurun_out/pmc_c.log"
echo DONE
