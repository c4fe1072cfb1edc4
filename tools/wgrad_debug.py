#!/usr/bin/env python3
"""Pin down the ChunkedLinear backward slowdown layer by layer."""
import os, sys
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
from ray_shuffling_data_loader_amd.models.mlp import _wgrad_chunks

def t(fn, iters=20, warmup=3):
    for _ in range(warmup): fn()
    torch.cuda.synchronize()
    a = torch.cuda.Event(enable_timing=True); b = torch.cuda.Event(enable_timing=True)
    a.record()
    for _ in range(iters): fn()
    b.record(); torch.cuda.synchronize()
    return a.elapsed_time(b) / iters

def main():
    dev = "cuda"; M = 250_000; dt = torch.bfloat16
    shapes = [(512, 100), (256, 512), (128, 256), (1, 128)]
    for N, K in shapes:
        dy = torch.randn(M, N, device=dev, dtype=dt)
        x = torch.randn(M, K, device=dev, dtype=dt)
        w = torch.randn(N, K, device=dev, dtype=dt)
        c = _wgrad_chunks(M)
        ms_mm = t(lambda: dy.t() @ x)
        def bmm():
            return torch.bmm(dy.view(c, M//c, N).transpose(1,2), x.view(c, M//c, K)).sum(0)
        ms_bmm = t(bmm)
        ms_dx = t(lambda: dy @ w)
        print(f"L N={N:4d} K={K:4d}: wgrad mm {ms_mm:7.3f}  bmm(c={c}) {ms_bmm:7.3f}  dgrad {ms_dx:7.3f} ms")




def kernel_bench():
    from ray_shuffling_data_loader_amd.ops.shuffle_ops import wgrad
    dev = "cuda"; M = 250_000; dt = torch.bfloat16
    for N, K in [(512, 100), (256, 512), (128, 256)]:
        dy = torch.randn(M, N, device=dev, dtype=dt)
        x = torch.randn(M, K, device=dev, dtype=dt)
        ms = t(lambda: wgrad(dy, x, True), iters=20)
        by = (M * N + M * K) * 2
        print(f"wgrad kernel N={N:3d} K={K:3d}: {ms:7.3f} ms  "
              f"{by/ms/1e6:6.0f} GB/s")
        c = 16
        ms2 = t(lambda: (torch.bmm(dy.view(c, M//c, N).transpose(1, 2),
                                   x.view(c, M//c, K)).sum(0), dy.sum(0)))
        print(f"  bmm+sum reference    : {ms2:7.3f} ms")


import os as _os
if _os.environ.get("RSDL_WGRAD_BENCH") == "1":
    kernel_bench()
    raise SystemExit


if __name__ == "__main__" and _os.environ.get("RSDL_WGRAD_BENCH") != "1":
    main()
