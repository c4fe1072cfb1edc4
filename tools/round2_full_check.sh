#!/usr/bin/env bash
set -x
cd "$(dirname "$0")/.."
REPO="$PWD"
mkdir -p gpurun_out
# 1. FULL gpu pytest (incl. new shm-transport test, tightened wgrad tol).
timeout 600 python -m pytest tests -m gpu -q > gpurun_out/pytest_gpu.log 2>&1
tail -3 gpurun_out/pytest_gpu.log
# 2. Chain numerics (experimental gate).
RSDL_EXPERIMENTAL=1 timeout 240 python -m pytest tests/test_gpu_kernels.py -m gpu -q \
    -k "fwd_chain or bwd_chain or fused_step" 2>&1 | tail -2
# 3. Driver-flag bench.
timeout 420 python bench.py --gpus 1 --steps 20 --warmup 5 2> gpurun_out/bench_full.err | tee gpurun_out/bench_full.json
# 4. Fresh per-kernel stats + instruction PMC.
export TMPDIR=/tmp
cd /tmp
RSDL_PROF_ITERS=10 timeout 300 rocprofv3 --kernel-trace --stats -d "$REPO/gpurun_out/prof_fused" -- \
    python "$REPO/tools/profile_fused_step.py" 2>&1 | tail -1
RSDL_PROF_ITERS=5 timeout 300 rocprofv3 --pmc SQ_INSTS_LDS SQ_INSTS_MFMA SQ_INSTS_VALU SQ_INSTS_VMEM \
  --kernel-trace --stats -d "$REPO/gpurun_out/pmc_b" -- \
  python "$REPO/tools/profile_fused_step.py" > "$REPO/gpurun_out/pmc_b.log" 2>&1 || tail -3 "$REPO/gpurun_out/pmc_b.log"
echo DONE
