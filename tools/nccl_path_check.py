#!/usr/bin/env python3
"""Exercise the RCCL code path of the shuffle fabric on a single GPU
(world=1 self-exchange): validates nccl-backend argument handling
(uint8 all_to_all_single splits, device count tensors, eager shuffle-group
init) without needing multiple GPUs. Launch under torchrun nproc=1."""
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch
import torch.distributed as dist

from ray_shuffling_data_loader_amd.parallel import fabric


def main():
    world, rank = fabric.init_from_env(backend="nccl")
    assert world == 1 and dist.get_backend() == "nccl"
    dev = torch.device("cuda", 0)
    g = fabric.get_shuffle_group()  # eager nccl comm init on main thread

    n, stride = 100_000, 416
    rows = torch.randint(0, 256, (n, stride), dtype=torch.uint8, device=dev)
    counts = torch.tensor([n], device=dev)
    recv, rc = fabric.exchange_rows(rows, counts, g)
    assert recv.shape == rows.shape
    assert torch.equal(recv, rows), "self-exchange must be identity"
    assert rc.tolist() == [n]
    print("NCCL PATH CHECK: PASS (uint8 all_to_all_single, counts exchange,"
          " shuffle group init)")
    dist.destroy_process_group()


if __name__ == "__main__":
    main()
