#!/usr/bin/env python3
"""End-to-end DDP training example over the MI355X shuffling data loader.

Counterpart of the reference's Horovod example (reference:
examples/horovod/ray_torch_shuffle.py): every rank builds a
TorchShufflingDataset (rank 0 / symmetric engines kick off the shuffle),
per-step **batch wait times** are measured exactly like the reference
(ray_torch_shuffle.py:195-231 — the p50 batch-wait north star), and
gradients are synchronized with RCCL allreduce via torch DDP instead of
Horovod NCCL (reference N9/N10). Unlike the reference — whose train step is
`time.sleep(mock_train_step_time)` with the real fwd/bwd commented out
(ray_torch_shuffle.py:209-218) — the step here runs a real model; pass
--mock-train-step-time to reproduce the reference's sleep-based methodology
instead.

Launch:
  python -m torch.distributed.run --nnodes=1 --nproc-per-node 8 \
      --master-addr 127.0.0.1 examples/ddp/train_ddp.py
Single process (no torchrun) also works: local central mode.
"""

import argparse
import os

import sys
import tempfile
import time

sys.path.insert(
    0,
    os.path.dirname(
        os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    ),
)

import numpy as np
import torch

from ray_shuffling_data_loader_amd.data_generation import (
    float_data_spec,
    generate_data,
)
from ray_shuffling_data_loader_amd.models.mlp import TabularMLP
from ray_shuffling_data_loader_amd.parallel import fabric
from ray_shuffling_data_loader_amd.torch_dataset import TorchShufflingDataset


def parse_args():
    p = argparse.ArgumentParser()
    p.add_argument("--num-rows", type=int, default=2 * 10**7)
    p.add_argument("--num-files", type=int, default=25)
    p.add_argument("--num-row-groups-per-file", type=int, default=5)
    p.add_argument("--num-cols", type=int, default=100)
    p.add_argument("--batch-size", type=int, default=250_000)
    p.add_argument("--num-epochs", type=int, default=10)
    p.add_argument("--num-reducers", type=int, default=32)
    p.add_argument("--max-concurrent-epochs", type=int, default=2)
    p.add_argument("--mock-train-step-time", type=float, default=None,
                   help="replace the real fwd/bwd with sleep(T) "
                   "(the reference example's methodology)")
    p.add_argument("--spec", choices=["float", "tabular"], default="float",
                   help="float: 100 fp32 cols + TabularMLP; tabular: the "
                   "reference DATA_SPEC (19 int64 categoricals) + "
                   "EmbeddingMLP")
    p.add_argument("--data-dir", type=str, default=None)
    p.add_argument("--cache-files", action="store_true")
    p.add_argument("--lr", type=float, default=1e-3)
    p.add_argument("--fp16-allreduce", action="store_true",
                   help="compress DDP gradients to fp16 for the RCCL "
                   "allreduce (reference ray_torch_shuffle.py:184-185 "
                   "hvd fp16 compression parity)")
    p.add_argument("--device", type=str, default=None)
    return p.parse_args()


def train_main(args, world, rank):
    if args.device:
        device = torch.device(args.device)
    elif torch.cuda.is_available():
        device = torch.device("cuda", rank % torch.cuda.device_count())
        torch.cuda.set_device(device)
    else:
        device = torch.device("cpu")

    from ray_shuffling_data_loader_amd.data_generation import DATA_SPEC
    from ray_shuffling_data_loader_amd.models.mlp import EmbeddingMLP

    tabular = args.spec == "tabular"
    data_dir = args.data_dir or os.path.join(
        tempfile.gettempdir(), f"rsdl_ddp_example_data_{args.spec}"
    )
    spec = DATA_SPEC if tabular else float_data_spec(args.num_cols)
    if rank == 0 and (
        not args.cache_files
        or not os.path.isdir(data_dir)
        or not os.listdir(data_dir)
    ):
        print(f"Generating {args.num_rows} rows over {args.num_files} files")
        generate_data(
            args.num_rows,
            args.num_files,
            args.num_row_groups_per_file,
            0.0,
            data_dir,
            spec=spec,
            include_key=not tabular,
        )
    if world > 1:
        torch.distributed.barrier()
    filenames = sorted(
        os.path.join(data_dir, f) for f in os.listdir(data_dir)
    )

    if tabular:
        # Reference example shape: 19 int64 categorical columns + fp64
        # label (ray_torch_shuffle.py:256-278 over DATA_SPEC).
        feature_columns = [c for c in DATA_SPEC if c != "labels"]
        cardinalities = [DATA_SPEC[c][1] for c in feature_columns]
        ds = TorchShufflingDataset(
            filenames,
            args.num_epochs,
            num_trainers=world,
            batch_size=args.batch_size,
            rank=rank,
            num_reducers=args.num_reducers,
            max_concurrent_epochs=args.max_concurrent_epochs,
            feature_columns=feature_columns,
            feature_types=[torch.int64] * len(feature_columns),
            label_column="labels",
            device=device,
        )
        model = EmbeddingMLP(cardinalities).to(device)
    else:
        feature_columns = [f"f{i}" for i in range(args.num_cols)]
        ds = TorchShufflingDataset(
            filenames,
            args.num_epochs,
            num_trainers=world,
            batch_size=args.batch_size,
            rank=rank,
            num_reducers=args.num_reducers,
            max_concurrent_epochs=args.max_concurrent_epochs,
            feature_columns=feature_columns,
            label_column="labels",
            feature_matrix=True,
            device=device,
        )
        model = TabularMLP(args.num_cols).to(device)
    if world > 1:
        model = torch.nn.parallel.DistributedDataParallel(model)
        if args.fp16_allreduce:
            from torch.distributed.algorithms.ddp_comm_hooks import (
                default_hooks,
            )

            model.register_comm_hook(
                None, default_hooks.fp16_compress_hook
            )
    opt = torch.optim.SGD(model.parameters(), lr=args.lr, momentum=0.9)
    loss_fn = torch.nn.MSELoss()

    import contextlib

    for epoch in range(args.num_epochs):
        ds.set_epoch(epoch)
        wait_times = []
        n_batches = 0
        t_epoch = time.perf_counter()
        t_wait = time.perf_counter()
        # Reducer partitions are binomial, so per-rank batch counts differ;
        # DDP's join() shadows the missing allreduces on ranks that finish
        # the epoch early (without it the gradient allreduce deadlocks —
        # the reference example only avoids this because its train step is
        # a sleep with the real fwd/bwd commented out).
        join_ctx = (
            model.join()
            if world > 1 and args.mock_train_step_time is None
            else contextlib.nullcontext()
        )
        with join_ctx:
            for data, target in ds:
                wait_times.append(time.perf_counter() - t_wait)
                if args.mock_train_step_time is not None:
                    time.sleep(args.mock_train_step_time)
                else:
                    x = data if tabular else data[0]
                    opt.zero_grad(set_to_none=True)
                    loss = loss_fn(model(x), target)
                    loss.backward()
                    opt.step()
                n_batches += 1
                t_wait = time.perf_counter()
        if device.type == "cuda":
            torch.cuda.synchronize(device)
        dur = time.perf_counter() - t_epoch
        # Batch-wait statistics: the reference's per-epoch printout
        # (ray_torch_shuffle.py:221-230).
        wt = np.array(wait_times)
        print(
            f"[rank {rank}] epoch {epoch}: {n_batches} batches in "
            f"{dur:.2f}s | batch wait mean {wt.mean():.4f}s std "
            f"{wt.std():.4f} p50 {np.percentile(wt, 50):.4f} max "
            f"{wt.max():.4f} min {wt.min():.4f}",
            flush=True,
        )
    if rank == 0:
        print("Done training.")


def main():
    args = parse_args()
    world, rank = fabric.init_from_env()
    train_main(args, world, rank)
    if world > 1:
        torch.distributed.destroy_process_group()


if __name__ == "__main__":
    main()
