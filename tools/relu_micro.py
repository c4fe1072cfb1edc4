import os, sys
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch, time
from ray_shuffling_data_loader_amd.ops import shuffle_ops
hip = shuffle_ops._load_hip()
def t(f, n=50):
    for _ in range(10): f()
    torch.cuda.synchronize(); t0 = time.perf_counter()
    for _ in range(n): f()
    torch.cuda.synchronize(); return (time.perf_counter()-t0)/n*1e3
for m, n in [(250_000, 512), (250_000, 256), (250_000, 128)]:
    dy = torch.randn(m, n, device="cuda").bfloat16()
    y = (torch.randn(m, n, device="cuda") - 0.3).bfloat16()
    fused = t(lambda: hip.relu_bwd_bias(dy, y))
    unf = t(lambda: (torch.ops.aten.threshold_backward(dy, y, 0).sum(0)))
    print(f"{m}x{n}: fused {fused:.3f} ms  unfused(tb+sum) {unf:.3f} ms")
