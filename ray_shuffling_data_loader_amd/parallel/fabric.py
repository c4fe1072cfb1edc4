"""Process fabric + collective row exchange.

The reference's bulk data plane is the Ray object store (map tasks return
per-reducer DataFrames, reduce tasks fetch their column — the implicit
all-to-all at reference shuffle.py:112-123). Here the fabric is one process
per GPU under ``torch.distributed`` and the exchange is an explicit
**RCCL all-to-all over xGMI** (`all_to_all_single` on the "nccl" backend,
which IS RCCL on ROCm): xGMI is 7 point-to-point links per GPU, and
all-to-all drives all 7 concurrently, unlike a ring collective that is
single-link bound (SURVEY.md §5 'Distributed communication backend').

On the CPU/gloo backend (multi-process tests, no GPU) the same exchange runs
as paired isend/irecv, since gloo lacks all_to_all.
"""

import datetime
import os
from typing import Optional, Tuple

import torch
import torch.distributed as dist


def dist_info() -> Tuple[int, int, bool]:
    """(world_size, rank, is_initialized)."""
    if dist.is_available() and dist.is_initialized():
        return dist.get_world_size(), dist.get_rank(), True
    return 1, 0, False


def init_from_env(
    backend: Optional[str] = None,
    timeout_s: float = 1800.0,
) -> Tuple[int, int]:
    """Initialize torch.distributed from torchrun env vars if present.
    Returns (world_size, rank). No-op outside a distributed launch."""
    if dist.is_initialized():
        return dist.get_world_size(), dist.get_rank()
    if "RANK" not in os.environ or "WORLD_SIZE" not in os.environ:
        return 1, 0
    if backend is None:
        backend = "nccl" if torch.cuda.is_available() else "gloo"
    if backend == "nccl":
        local_rank = int(os.environ.get("LOCAL_RANK", os.environ["RANK"]))
        torch.cuda.set_device(local_rank % max(1, torch.cuda.device_count()))
    dist.init_process_group(
        backend=backend, timeout=datetime.timedelta(seconds=timeout_s)
    )
    return dist.get_world_size(), dist.get_rank()


_shuffle_group = None


def get_shuffle_group():
    """A dedicated communicator for the shuffle engine's collectives.

    The engine runs on a background thread while DDP allreduces run on the
    trainer's main thread; if both used the default process group their
    collectives could interleave in different orders on different ranks
    (undefined behavior / deadlock). A separate group gives the shuffle its
    own NCCL/gloo communicator, within which its collectives are strictly
    epoch-ordered. Must first be called from the MAIN thread on all ranks
    in the same order (dataset construction does this)."""
    global _shuffle_group
    if _shuffle_group is None:
        _shuffle_group = dist.new_group(
            ranks=list(range(dist.get_world_size()))
        )
        # Eagerly initialize the communicator HERE on the main thread, so
        # its (collective) init can never race the default communicator's
        # first DDP use from another thread.
        probe = torch.zeros(1)
        if dist.get_backend(_shuffle_group) == "nccl":
            probe = probe.cuda()
        dist.all_reduce(probe, group=_shuffle_group)
    return _shuffle_group


def exchange_counts(
    send_counts: torch.Tensor, group=None, device=None
) -> torch.Tensor:
    """Size-exchange phase: every rank contributes its [world] row counts per
    destination; returns this rank's [world] receive counts (rows arriving
    from each source). Needed because reducer partition sizes are binomial
    (SURVEY.md §7 hard part (b))."""
    world = dist.get_world_size(group)
    rank = dist.get_rank(group)
    backend = dist.get_backend(group)
    send_counts = send_counts.to(torch.long)
    if backend == "nccl":
        # RCCL collectives need device tensors.
        send_counts = send_counts.to(device or "cuda")
    else:
        send_counts = send_counts.cpu()
    gathered = [torch.zeros_like(send_counts) for _ in range(world)]
    dist.all_gather(gathered, send_counts, group=group)
    return torch.stack(gathered)[:, rank].cpu().contiguous()


def exchange_rows(
    grouped: torch.Tensor,
    send_counts: torch.Tensor,
    group=None,
) -> Tuple[torch.Tensor, torch.Tensor]:
    """All-to-all of packed rows.

    ``grouped``: [N, row_stride] rows sorted by destination rank;
    ``send_counts``: rows per destination ([world]). Returns
    (received rows [M, row_stride] grouped by source rank, recv_counts).
    """
    world = dist.get_world_size(group)
    rank = dist.get_rank(group)
    recv_counts = exchange_counts(send_counts, group, device=grouped.device)
    in_splits = send_counts.cpu().tolist()  # one sync, not world syncs
    out_splits = recv_counts.tolist()
    recv = torch.empty(
        (sum(out_splits), grouped.shape[1]),
        dtype=grouped.dtype,
        device=grouped.device,
    )
    backend = dist.get_backend(group)
    if backend == "nccl":
        # RCCL all-to-all over xGMI: traffic spread over all 7 p2p links.
        dist.all_to_all_single(
            recv, grouped.contiguous(), out_splits, in_splits, group=group
        )
        return recv, recv_counts
    # gloo fallback: paired non-blocking send/recv (all_to_all unsupported).
    in_offs = [0]
    for c in in_splits:
        in_offs.append(in_offs[-1] + c)
    out_offs = [0]
    for c in out_splits:
        out_offs.append(out_offs[-1] + c)
    grouped = grouped.contiguous()
    # Self copy without the wire.
    if out_splits[rank]:
        recv[out_offs[rank] : out_offs[rank + 1]] = grouped[
            in_offs[rank] : in_offs[rank + 1]
        ]
    reqs = []
    for peer in range(world):
        if peer == rank:
            continue
        if out_splits[peer]:
            reqs.append(
                dist.irecv(
                    recv[out_offs[peer] : out_offs[peer + 1]],
                    src=peer,
                    group=group,
                )
            )
    for peer in range(world):
        if peer == rank:
            continue
        if in_splits[peer]:
            reqs.append(
                dist.isend(
                    grouped[in_offs[peer] : in_offs[peer + 1]].contiguous(),
                    dst=peer,
                    group=group,
                )
            )
    for r in reqs:
        r.wait()
    return recv, recv_counts
