#!/usr/bin/env bash
# Round-3 first GPU call: regression + A/B of the statically-verified
# pinned-schedule wgrad/slab_reduce variants built at the end of round 2
# (profiles/r02/wgrad_sched_asm.md — schedule verified by disassembly,
# never yet run on hardware).
# Usage: /usr/local/graft/bin/gpurun --timeout 1500 -- 'bash tools/round3_gpu_validation.sh'
set -x
cd "$(dirname "$0")/.."
mkdir -p gpurun_out

# 1. Regression: full GPU suite at HEAD (baseline kernels).
timeout 420 python -m pytest tests -m gpu -q 2>&1 | tail -3 | tee gpurun_out/r3_suite_base.txt

# 2. Parity of the SCHED variants (same oracles, env flipped).
RSDL_WGRAD_SCHED=1 timeout 300 python -m pytest tests/test_gpu_kernels.py -m gpu -q \
    -k "wgrad_frag or fused_step or chain" 2>&1 | tail -3 | tee gpurun_out/r3_suite_sched.txt

# 3. Kernel-level A/B: wgrad_frag + slab_reduce times, both schedules.
for S in 0 1; do
  RSDL_WGRAD_SCHED=$S timeout 240 \
    python tools/wgrad_frag_bench.py 2>&1 | tail -6 | tee gpurun_out/r3_wgrad_sched$S.txt
done

# 4. Step-level A/B at the flagship shape.
for S in 0 1; do
  RSDL_WGRAD_SCHED=$S timeout 240 python tools/profile_fused_step.py 2>&1 \
    | tail -6 | tee gpurun_out/r3_step_sched$S.txt
done

# 5. Driver-flag bench A/B (one run each; box-to-box band is ~±2%).
for S in 0 1; do
  RSDL_WGRAD_SCHED=$S timeout 180 python bench.py --gpus 1 --steps 20 --warmup 5 \
    > gpurun_out/r3_bench_sched$S.json 2>gpurun_out/r3_bench_sched$S.err
done
tail -1 gpurun_out/r3_bench_sched0.json
tail -1 gpurun_out/r3_bench_sched1.json
