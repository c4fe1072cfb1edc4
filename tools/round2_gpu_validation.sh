#!/usr/bin/env bash
# Round-2 first GPU call: regression + everything round 1 could not validate.
# Usage: /usr/local/graft/bin/gpurun --timeout 1500 -- 'bash tools/round2_gpu_validation.sh'
set -x
cd "$(dirname "$0")/.."
mkdir -p gpurun_out

# 1. Regression: round-1 validated surface + round-2 CPU-side changes
#    (queue close, shm transport default, row-group ingest, new wgrad tol).
timeout 420 python -m pytest tests -m gpu -q 2>&1 | tail -4

# 2. The experimental chain kernels (index math is CPU-verified in
#    tests/test_chain_sim.py; this is their first hardware run).
RSDL_EXPERIMENTAL=1 timeout 240 python -m pytest tests/test_gpu_kernels.py -m gpu -q \
    -k "fwd_chain or bwd_chain or fused_step" 2>&1 | tail -3

# 3. Fused-chain step vs eager step timing.
timeout 240 python - <<'PY'
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

# 4. Uncached ingest rate with row-group-parallel tasks (8 RGs/file now).
timeout 300 python - <<'PY'
import os, tempfile, time, torch
from ray_shuffling_data_loader_amd.data_generation import float_data_spec, generate_file
from ray_shuffling_data_loader_amd.io import infer_schema, read_files_packed
from concurrent.futures import ThreadPoolExecutor

spec = float_data_spec(100)
d = tempfile.mkdtemp()
rows_per_file = 3_125_000
with ThreadPoolExecutor(8) as p:
    list(p.map(lambda i: generate_file(i, i*rows_per_file, rows_per_file,
        max(1, rows_per_file // 390_625), d, spec=spec, include_key=False), range(4)))
fns = sorted(os.path.join(d, f) for f in os.listdir(d))
schema = infer_schema(fns[0])
dev = torch.device("cuda", 0)
for threads in (8, 16):
    # warm page cache on first pass; report second
    for rep in range(2):
        torch.cuda.synchronize(); t0 = time.perf_counter()
        packed = read_files_packed(fns, schema, dev, reader_threads=threads)
        torch.cuda.synchronize(); dt = time.perf_counter() - t0
        gb = packed.numel() / 1e9
        print(f"ingest threads={threads} rep={rep}: {dt:.2f}s  {gb/dt:.1f} GB/s ({packed.shape[0]} rows)")
    del packed
PY

# 5. bench.py with the driver's exact flags: verify the rollover lands in
#    the timed window and the number survives.
timeout 420 python bench.py --gpus 1 --steps 20 --warmup 5 2> gpurun_out/bench_r2_first.err | tee gpurun_out/bench_r2_first.json
tail -3 gpurun_out/bench_r2_first.err
