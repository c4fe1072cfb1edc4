#!/usr/bin/env bash
# Round-2 first GPU call: validate everything round 1 could not.
# Usage: /usr/local/graft/bin/gpurun --timeout 900 -- 'bash tools/round2_gpu_validation.sh'
set -x
cd "$(dirname "$0")/.."

# 1. Regression: the round-1 validated surface still passes.
python -m pytest tests -m gpu -q 2>&1 | tail -2

# 2. The experimental chain kernels (index math is CPU-verified in
#    tests/test_chain_sim.py; this is their first hardware run).
RSDL_EXPERIMENTAL=1 python -m pytest tests/test_gpu_kernels.py -m gpu -q \
    -k "fwd_chain or bwd_chain or fused_step" 2>&1 | tail -2

# 3. If (2) passed: time the fused step vs the eager step.
python - <<'PY'
import time, torch
from ray_shuffling_data_loader_amd.models.fused_step import fused_step
from ray_shuffling_data_loader_amd.models.mlp import TabularMLP

M = 250_000
model = TabularMLP(100).cuda()
x = torch.randn(M, 100, device="cuda").bfloat16()
t = torch.randn(M, 1, device="cuda")
opt = torch.optim.SGD(model.parameters(), lr=1e-3, momentum=0.9, fused=True)

def eager():
    opt.zero_grad(set_to_none=True)
    with torch.autocast("cuda", torch.bfloat16):
        loss = torch.nn.functional.mse_loss(model(x).float(), t)
    loss.backward()
    opt.step()

def fused():
    fused_step(model, x, t)
    opt.step()

for name, fn in [("eager", eager), ("fused", fused)]:
    for _ in range(10):
        fn()
    torch.cuda.synchronize(); t0 = time.perf_counter()
    for _ in range(40):
        fn()
    torch.cuda.synchronize()
    print(f"{name}: {(time.perf_counter()-t0)/40*1e3:.3f} ms/step")
PY
