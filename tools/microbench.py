#!/usr/bin/env python3
"""GPU micro-benchmarks for the shuffle hot ops + trainer GEMM input layout.

Run on an MI355X box:  python tools/microbench.py
Writes a summary to stdout (capture into gpurun_out/)."""

import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch

from ray_shuffling_data_loader_amd.ops.shuffle_ops import (
    gather_rows,
    pack_columns,
    partition_rows,
    unpack_permute,
)
from ray_shuffling_data_loader_amd.utils.schema import ColumnSpec, Schema


def timeit_gpu(fn, iters=20, warmup=3):
    for _ in range(warmup):
        fn()
    torch.cuda.synchronize()
    t0 = torch.cuda.Event(enable_timing=True)
    t1 = torch.cuda.Event(enable_timing=True)
    t0.record()
    for _ in range(iters):
        fn()
    t1.record()
    torch.cuda.synchronize()
    return t0.elapsed_time(t1) / iters  # ms


def main():
    assert torch.cuda.is_available()
    dev = torch.device("cuda", 0)
    print(f"device: {torch.cuda.get_device_name(0)}")

    n = 2_500_000
    stride = 416  # 100 fp32 features + label + pad
    src = torch.randint(0, 256, (n, stride), dtype=torch.uint8, device=dev)
    perm64 = torch.randperm(n, device=dev)
    perm32 = perm64.to(torch.int32)
    out = torch.empty_like(src)

    ms = timeit_gpu(lambda: gather_rows(src, perm64, out=out))
    print(
        f"gather_rows i64 perm : {ms:7.3f} ms  "
        f"{2 * n * stride / ms / 1e6:7.0f} GB/s"
    )
    ms = timeit_gpu(lambda: gather_rows(src, perm32, out=out))
    print(
        f"gather_rows i32 perm : {ms:7.3f} ms  "
        f"{2 * n * stride / ms / 1e6:7.0f} GB/s"
    )
    ms = timeit_gpu(lambda: out.copy_(src))
    print(
        f"plain copy (roofline): {ms:7.3f} ms  "
        f"{2 * n * stride / ms / 1e6:7.0f} GB/s"
    )

    # Fused unpack+permute into contiguous feature matrix + labels.
    schema = Schema(
        [
            ColumnSpec("features", torch.float32, 100),
            ColumnSpec("labels", torch.float32, 1),
        ]
    )
    packed = src  # reinterpret (bytes are random, fine for perf)
    ms = timeit_gpu(
        lambda: unpack_permute(packed, schema, perm=perm64), iters=10
    )
    bytes_moved = n * (schema.payload_bytes + stride)  # read row, write cols
    print(
        f"unpack_permute fused : {ms:7.3f} ms  "
        f"{bytes_moved / ms / 1e6:7.0f} GB/s"
    )
    ms = timeit_gpu(
        lambda: unpack_permute(
            packed,
            schema,
            perm=perm64,
            out_dtypes={"features": torch.bfloat16},
        ),
        iters=10,
    )
    bytes_bf16 = n * (stride + 100 * 2 + 4)
    print(
        f"unpack_permute ->bf16: {ms:7.3f} ms  "
        f"{bytes_bf16 / ms / 1e6:7.0f} GB/s"
    )

    # partition_rows (fused hist+scan+scatter+gather) for 8 destinations
    dest = torch.randint(0, 8, (n,), device=dev, dtype=torch.int32)
    ms = timeit_gpu(lambda: partition_rows(src, dest, 8), iters=10)
    print(f"partition_rows 8-way : {ms:7.3f} ms")

    def part_sort():
        order = torch.argsort(dest.long(), stable=True)
        return gather_rows(src, order)

    ms = timeit_gpu(part_sort, iters=10)
    print(f"partition via argsort: {ms:7.3f} ms")

    # Tiled pack: 101 fp32 scalar columns -> packed rows (map side).
    from ray_shuffling_data_loader_amd.utils.schema import Schema as Sch

    pcols = {f"c{i}": torch.randn(n, device=dev) for i in range(101)}
    psch = Sch(
        [ColumnSpec(f"c{i}", torch.float32, 1) for i in range(101)]
    )
    ms = timeit_gpu(lambda: pack_columns(pcols, psch), iters=10)
    pbytes = n * (101 * 4 + psch.row_stride)
    print(
        f"pack_columns tiled   : {ms:7.3f} ms  "
        f"{pbytes / ms / 1e6:7.0f} GB/s"
    )

    if os.environ.get("RSDL_MB_KERNELS_ONLY") == "1":
        return

    # GEMM input layout A/B: [250k,100] fp32 strided (lda=104) vs contiguous
    b = 250_000
    lin = torch.nn.Linear(100, 512).to(dev)
    xs_raw = torch.randn(b, 104, device=dev)
    x_strided = xs_raw[:, :100]
    x_contig = x_strided.contiguous()
    ms = timeit_gpu(lambda: lin(x_strided), iters=30)
    print(f"Linear(100->512) strided A : {ms:7.3f} ms")
    ms = timeit_gpu(lambda: lin(x_contig), iters=30)
    print(f"Linear(100->512) contig  A : {ms:7.3f} ms")

    # Full fwd+bwd step A/B
    from ray_shuffling_data_loader_amd.models.mlp import TabularMLP

    model = TabularMLP(100).to(dev)
    opt = torch.optim.SGD(model.parameters(), lr=1e-3, momentum=0.9)
    y = torch.randn(b, 1, device=dev)

    def step(x):
        opt.zero_grad(set_to_none=True)
        loss = torch.nn.functional.mse_loss(model(x), y)
        loss.backward()
        opt.step()

    ms = timeit_gpu(lambda: step(x_strided), iters=20)
    print(f"MLP step strided input     : {ms:7.3f} ms")
    ms = timeit_gpu(lambda: step(x_contig), iters=20)
    print(f"MLP step contig input      : {ms:7.3f} ms")

    x_bf16 = x_contig.to(torch.bfloat16)

    def step_amp(x):
        opt.zero_grad(set_to_none=True)
        with torch.autocast("cuda", torch.bfloat16):
            out = model(x)
            loss = torch.nn.functional.mse_loss(out.float(), y)
        loss.backward()
        opt.step()

    ms = timeit_gpu(lambda: step_amp(x_bf16), iters=20)
    print(f"MLP step bf16 autocast     : {ms:7.3f} ms")

    model_bf = TabularMLP(100).to(dev).to(torch.bfloat16)
    opt_bf = torch.optim.SGD(model_bf.parameters(), lr=1e-3, momentum=0.9)

    def step_full_bf16(x):
        opt_bf.zero_grad(set_to_none=True)
        loss = torch.nn.functional.mse_loss(model_bf(x).float(), y)
        loss.backward()
        opt_bf.step()

    ms = timeit_gpu(lambda: step_full_bf16(x_bf16), iters=20)
    print(f"MLP step pure bf16 model   : {ms:7.3f} ms")

    # Step decomposition (bf16 autocast path)
    with torch.autocast("cuda", torch.bfloat16):
        out_f = model(x_bf16)
        loss_f = torch.nn.functional.mse_loss(out_f.float(), y)

    def fwd_only():
        with torch.autocast("cuda", torch.bfloat16):
            out = model(x_bf16)
            return torch.nn.functional.mse_loss(out.float(), y)

    ms = timeit_gpu(fwd_only, iters=20)
    print(f"  fwd+loss only            : {ms:7.3f} ms")

    def fwd_bwd():
        opt.zero_grad(set_to_none=True)
        loss = fwd_only()
        loss.backward()

    ms = timeit_gpu(fwd_bwd, iters=20)
    print(f"  fwd+bwd (no opt)         : {ms:7.3f} ms")

    gparams = [p for p in model.parameters()]

    def opt_only():
        opt.step()

    ms = timeit_gpu(opt_only, iters=20)
    print(f"  opt.step only            : {ms:7.3f} ms")


if __name__ == "__main__":
    main()
