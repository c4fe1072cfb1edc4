"""Multi-process distributed shuffle tests (gloo backend, world_size=2, CPU).

Covers the symmetric collective path that becomes the RCCL/xGMI flagship on
GPU: per-rank engines, destination-rank assignment, size exchange +
all-to-all rows exchange (isend/irecv on gloo), per-rank reducer split, and
the exactly-once global row invariant.
"""

import os
import tempfile

import pytest
import torch


def _run_rank(rank, world, init_method, filenames, num_rows, result_q,
              num_epochs):
    try:
        import torch.distributed as dist

        dist.init_process_group(
            "gloo", init_method=init_method, rank=rank, world_size=world
        )
        from ray_shuffling_data_loader_amd.dataset import ShufflingDataset

        ds = ShufflingDataset(
            filenames,
            num_epochs,
            num_trainers=world,
            batch_size=1000,
            rank=rank,
            num_reducers=4,
            seed=42,
        )
        out = {}
        for epoch in range(num_epochs):
            ds.set_epoch(epoch)
            keys = [b["key"] for b in ds]
            out[epoch] = (
                torch.cat(keys).tolist() if keys else []
            )
        result_q.put((rank, out))
        dist.destroy_process_group()
    except Exception as e:  # surface to the parent
        import traceback

        result_q.put((rank, f"ERROR: {e}\n{traceback.format_exc()}"))


@pytest.mark.parametrize("num_epochs", [2])
def test_distributed_exactly_once(tmp_path, num_epochs, mp_spawn_context):
    from ray_shuffling_data_loader_amd.data_generation import generate_data

    num_rows = 12000
    filenames, _ = generate_data(num_rows, 4, 1, 0.0, str(tmp_path))
    filenames = list(filenames)

    world = 2
    port_file = tempfile.NamedTemporaryFile(delete=False)
    init_method = f"file://{port_file.name}"
    os.unlink(port_file.name)

    ctx = mp_spawn_context
    result_q = ctx.Queue()
    procs = [
        ctx.Process(
            target=_run_rank,
            args=(
                r,
                world,
                init_method,
                filenames,
                num_rows,
                result_q,
                num_epochs,
            ),
        )
        for r in range(world)
    ]
    for p in procs:
        p.start()
    results = {}
    for _ in range(world):
        rank, out = result_q.get(timeout=180)
        assert not isinstance(out, str), out
        results[rank] = out
    for p in procs:
        p.join(timeout=60)
        assert p.exitcode == 0

    for epoch in range(num_epochs):
        all_keys = sorted(results[0][epoch] + results[1][epoch])
        assert all_keys == list(range(num_rows)), (
            f"epoch {epoch}: global exactly-once violated "
            f"({len(all_keys)} rows)"
        )
        # Both ranks must actually receive a share (binomial around 1/2).
        assert len(results[0][epoch]) > num_rows // 4
        assert len(results[1][epoch]) > num_rows // 4
    # Different epochs produce different per-rank orderings.
    assert results[0][0] != results[0][1]


def test_distributed_fewer_files_than_ranks(tmp_path, mp_spawn_context):
    # One Parquet file, two ranks: rank 1 maps zero files but still
    # participates in the exchange and receives ~half the rows.
    from ray_shuffling_data_loader_amd.data_generation import generate_data

    num_rows = 6000
    filenames, _ = generate_data(num_rows, 1, 1, 0.0, str(tmp_path))
    filenames = list(filenames)

    port_file = tempfile.NamedTemporaryFile(delete=False)
    init_method = f"file://{port_file.name}"
    os.unlink(port_file.name)

    ctx = mp_spawn_context
    result_q = ctx.Queue()
    procs = [
        ctx.Process(
            target=_run_rank,
            args=(r, 2, init_method, filenames, num_rows, result_q, 1),
        )
        for r in range(2)
    ]
    for p in procs:
        p.start()
    results = {}
    for _ in range(2):
        rank, out = result_q.get(timeout=180)
        assert not isinstance(out, str), out
        results[rank] = out
    for p in procs:
        p.join(timeout=60)
        assert p.exitcode == 0
    all_keys = sorted(results[0][0] + results[1][0])
    assert all_keys == list(range(num_rows))
    assert len(results[1][0]) > num_rows // 4  # rank 1 still gets its share
