#!/usr/bin/env python3
"""Probe hipBLASLt on the exact MLP-step GEMM shapes (b=250k) in different
layouts, to see where the ~700us kernels come from and what layout/tuning
recovers roofline."""
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch


def t(fn, iters=30, warmup=5):
    for _ in range(warmup):
        fn()
    torch.cuda.synchronize()
    a = torch.cuda.Event(enable_timing=True)
    b = torch.cuda.Event(enable_timing=True)
    a.record()
    for _ in range(iters):
        fn()
    b.record()
    torch.cuda.synchronize()
    return a.elapsed_time(b) / iters


def main():
    dev = "cuda"
    M = 250_000
    dt = torch.bfloat16
    x = torch.randn(M, 100, device=dev, dtype=dt)
    W1 = torch.randn(512, 100, device=dev, dtype=dt)   # torch Linear layout
    W1t = W1.t().contiguous()                          # [100, 512]
    dy1 = torch.randn(M, 512, device=dev, dtype=dt)
    W2 = torch.randn(256, 512, device=dev, dtype=dt)
    dy2 = torch.randn(M, 256, device=dev, dtype=dt)

    def rep(name, ms, flops, bytes_):
        print(f"{name:<34} {ms:8.3f} ms  {flops/ms/1e9:7.1f} TF/s  "
              f"{bytes_/ms/1e6:7.0f} GB/s")

    f1 = 2 * M * 100 * 512
    b1 = (M * 100 + M * 512) * 2
    rep("fwd L1  x @ W1.T   (NT)", t(lambda: x @ W1.T), f1, b1)
    rep("fwd L1  x @ W1t    (NN)", t(lambda: x @ W1t), f1, b1)
    rep("fwd L1  linear(x,W1)   ", t(lambda: torch.nn.functional.linear(x, W1)), f1, b1)
    f2 = 2 * M * 512 * 256
    b2 = (M * 512 + M * 256) * 2
    rep("fwd L2  dy1 @ W2.T (NT)", t(lambda: dy1 @ W2.T), f2, b2)
    rep("fwd L2  dy1 @ W2t  (NN)", t(lambda: dy1 @ W2.t().contiguous()), f2, b2)
    # dgrad: dx = dy @ W
    rep("dgrad L1 dy1 @ W1  (NN)", t(lambda: dy1 @ W1), f1, b1)
    # wgrad: dW = dy.T @ x  (K = 250k)
    bw = (M * 512 + M * 100) * 2
    rep("wgrad L1 dy1.T @ x (TN)", t(lambda: dy1.T @ x), f1, bw)
    rep("wgrad L1 dy1.t()@x cont", t(lambda: dy1.t().contiguous() @ x), f1, bw)
    # fp32 accumulate out
    rep("fwd L1 NT out fp32", t(lambda: torch.matmul(x.float(), W1.float().T)), f1, b1 * 2)
    # smaller batch splits (64k)
    xs = x[:62500]
    rep("fwd L1 NT b=62.5k", t(lambda: xs @ W1.T), f1 / 4, b1 / 4)




def wgrad_variants():
    dev = "cuda"
    M, N, K = 250_000, 512, 100
    dt = torch.bfloat16
    dy = torch.randn(M, N, device=dev, dtype=dt)
    x = torch.randn(M, K, device=dev, dtype=dt)
    fl = 2 * M * N * K
    by = (M * N + M * K) * 2

    def rep(name, ms):
        print(f"{name:<34} {ms:8.3f} ms  {fl/ms/1e9:7.1f} TF/s  "
              f"{by/ms/1e6:7.0f} GB/s")

    rep("wgrad mm dy.T@x", t(lambda: dy.T @ x))
    for c in (4, 16, 64, 256):
        mc = M // c
        dyc = dy.view(c, mc, N)
        xc = x.view(c, mc, K)

        def f(dyc=dyc, xc=xc):
            return torch.bmm(dyc.transpose(1, 2), xc).sum(0)

        rep(f"wgrad bmm c={c}", t(f))
    # L2 shape
    N2, K2 = 256, 512
    dy2 = torch.randn(M, N2, device=dev, dtype=dt)
    y1 = torch.randn(M, K2, device=dev, dtype=dt)
    fl2 = 2 * M * N2 * K2
    by2 = (M * N2 + M * K2) * 2

    def rep2(name, ms):
        print(f"{name:<34} {ms:8.3f} ms  {fl2/ms/1e9:7.1f} TF/s  "
              f"{by2/ms/1e6:7.0f} GB/s")

    rep2("wgrad L2 mm", t(lambda: dy2.T @ y1))
    for c in (16, 64):
        mc = M // c

        def f2(c=c, mc=mc):
            return torch.bmm(
                dy2.view(c, mc, N2).transpose(1, 2), y1.view(c, mc, K2)
            ).sum(0)

        rep2(f"wgrad L2 bmm c={c}", t(f2))


if os.environ.get("RSDL_WGRAD") == "1":
    wgrad_variants()
    raise SystemExit


if __name__ == "__main__" and os.environ.get("RSDL_WGRAD") != "1":
    main()
