#!/usr/bin/env bash
# GPU call 2: chain-kernel numerics regression + timing + per-kernel profile.
set -x
cd "$(dirname "$0")/.."
REPO="$PWD"
mkdir -p gpurun_out

# 1. Numerics of the PIPELINED chain kernels.
RSDL_EXPERIMENTAL=1 timeout 240 python -m pytest tests/test_gpu_kernels.py -m gpu -q \
    -k "fwd_chain or bwd_chain or fused_step" > gpurun_out/chain_numerics.log 2>&1
tail -2 gpurun_out/chain_numerics.log

# 2. Timing: eager vs fused.
RSDL_PROF_MODE=eager timeout 240 python tools/profile_fused_step.py
RSDL_PROF_MODE=fused timeout 240 python tools/profile_fused_step.py

# 3. Per-kernel stats for the fused step.
export TMPDIR=/tmp
cd /tmp
timeout 300 rocprofv3 --kernel-trace --stats -d "$REPO/gpurun_out/prof_fused" -- \
    python "$REPO/tools/profile_fused_step.py" > "$REPO/gpurun_out/prof_fused.log" 2>&1
tail -3 "$REPO/gpurun_out/prof_fused.log"
# 4. And the eager step for comparison.
RSDL_PROF_MODE=eager timeout 300 rocprofv3 --kernel-trace --stats -d "$REPO/gpurun_out/prof_eager" -- \
    python "$REPO/tools/profile_fused_step.py" > "$REPO/gpurun_out/prof_eager.log" 2>&1
tail -3 "$REPO/gpurun_out/prof_eager.log"
grep -l . "$REPO"/gpurun_out/prof_fused/*/*stats* 2>/dev/null | head
