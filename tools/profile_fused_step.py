#!/usr/bin/env python3
"""Profiling driver: N iterations of the fused (or eager) train step at the
flagship shape, for rocprofv3 --stats runs. RSDL_PROF_MODE=eager|fused."""

import os
import sys
import time

sys.path.insert(
    0, os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
)

import torch  # noqa: E402

from ray_shuffling_data_loader_amd.models.fused_step import (  # noqa: E402
    fused_step,
)
from ray_shuffling_data_loader_amd.models.mlp import TabularMLP  # noqa: E402


def main():
    mode = os.environ.get("RSDL_PROF_MODE", "fused")
    iters = int(os.environ.get("RSDL_PROF_ITERS", "40"))
    M = int(os.environ.get("RSDL_PROF_M", "250000"))
    model = TabularMLP(100).cuda()
    x = torch.randn(M, 100, device="cuda").bfloat16()
    t = torch.randn(M, 1, device="cuda")
    opt = torch.optim.SGD(
        model.parameters(), lr=1e-3, momentum=0.9, fused=True
    )

    def eager():
        opt.zero_grad(set_to_none=True)
        with torch.autocast("cuda", torch.bfloat16):
            loss = torch.nn.functional.mse_loss(model(x).float(), t)
        loss.backward()
        opt.step()

    def fused():
        fused_step(model, x, t)
        opt.step()

    fn = fused if mode == "fused" else eager
    for _ in range(10):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    print(
        f"{mode}: {(time.perf_counter() - t0) / iters * 1e3:.3f} ms/step "
        f"(M={M}, iters={iters})"
    )


if __name__ == "__main__":
    main()
