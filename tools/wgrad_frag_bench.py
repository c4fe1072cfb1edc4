#!/usr/bin/env python3
"""Time the fragment-major wgrad path (wgrad_frag kernel + slab_reduce)
at the three flagship shapes. A/B the pinned-schedule variants with
RSDL_WGRAD_SCHED=1 (see profiles/r02/wgrad_sched_asm.md)."""
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch  # noqa: E402

from ray_shuffling_data_loader_amd.ops.shuffle_ops import wgrad_frag  # noqa: E402


def t(fn, iters=50, warmup=5):
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
    M = 250_000
    mchunks = 2 * ((M + 31) // 32)
    dev = "cuda"
    print(f"RSDL_WGRAD_SCHED={os.environ.get('RSDL_WGRAD_SCHED', '0')}")
    # (N, K) of dW; AT is dz^T [N-tiles][mchunks][512], BT is src^T.
    for N, K in [(512, 128), (256, 512), (128, 256)]:
        at = torch.randn(N // 32 * mchunks * 512, device=dev).bfloat16()
        bt = torch.randn(K // 32 * mchunks * 512, device=dev).bfloat16()
        ms = t(lambda: wgrad_frag(at, bt, N, K, mchunks))
        by = (at.numel() + bt.numel()) * 2  # bf16 stream bytes
        print(
            f"wgrad_frag N={N:3d} K={K:3d}: {ms * 1e3:7.1f} us  "
            f"{by / ms / 1e6:6.0f} GB/s effective"
        )


if __name__ == "__main__":
    main()
