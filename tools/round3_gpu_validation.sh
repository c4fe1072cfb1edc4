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

# 5. pi16 A/B: exchange-free chain epilogues (fwd -71 VALU/wave, bwd
#    -101; profiles/r02/wgrad_sched_asm.md + commit message).
RSDL_PI16=1 timeout 240 python tools/profile_fused_step.py 2>&1 \
  | tail -6 | tee gpurun_out/r3_step_pi16.txt

# 6. Driver-flag bench A/B over the three knob settings.
for CFG in "" "RSDL_WGRAD_SCHED=1" "RSDL_WGRAD_SCHED=1 RSDL_PI16=1" "RSDL_HIPRI_STEP=1"; do
  TAG=$(echo "$CFG" | tr -cd '01' | head -c 8)
  env $CFG timeout 180 python bench.py --gpus 1 --steps 20 --warmup 5 \
    > "gpurun_out/r3_bench_${TAG:-base}.json" 2>"gpurun_out/r3_bench_${TAG:-base}.err"
done
tail -1 gpurun_out/r3_bench_base.json
cat gpurun_out/r3_bench_*.json | tail -3
