"""The reference's non-distributed multi-process flow: rank 0 creates the
named queue + shuffle driver; a SEPARATE trainer process connects by name
and consumes its partition (reference dataset.py:52-84 + batch_queue
connect retry). CPU-only."""

import time

import torch


def _rank1_consumer(queue_name, filenames, num_rows, result_q):
    try:
        from ray_shuffling_data_loader_amd.dataset import ShufflingDataset

        ds = ShufflingDataset(
            filenames,
            num_epochs=1,
            num_trainers=2,
            batch_size=1000,
            rank=1,
            num_reducers=4,
            queue_name=queue_name,
        )
        ds.set_epoch(0)
        keys = [b["key"] for b in ds]
        result_q.put(torch.cat(keys).tolist() if keys else [])
    except Exception as e:
        import traceback

        result_q.put(f"ERROR: {e}\n{traceback.format_exc()}")


def test_connected_trainer_process(tmp_path, mp_spawn_context):
    from ray_shuffling_data_loader_amd.data_generation import generate_data
    from ray_shuffling_data_loader_amd.dataset import ShufflingDataset

    num_rows = 8000
    filenames, _ = generate_data(num_rows, 2, 1, 0.0, str(tmp_path))
    filenames = list(filenames)
    qname = f"mp_local_{time.time_ns()}"

    ctx = mp_spawn_context
    result_q = ctx.Queue()
    p = ctx.Process(
        target=_rank1_consumer,
        args=(qname, filenames, num_rows, result_q),
    )
    # Rank 0 first (creates the queue server); rank 1 retries connect.
    ds0 = ShufflingDataset(
        filenames,
        num_epochs=1,
        num_trainers=2,
        batch_size=1000,
        rank=0,
        num_reducers=4,
        queue_name=qname,
        seed=5,
    )
    p.start()
    ds0.set_epoch(0)
    keys0 = torch.cat([b["key"] for b in ds0]).tolist()
    out = result_q.get(timeout=120)
    assert not isinstance(out, str), out
    p.join(timeout=60)
    assert p.exitcode == 0
    all_keys = sorted(keys0 + out)
    assert all_keys == list(range(num_rows))


def test_connected_trainer_shm_transport(
    tmp_path, mp_spawn_context, monkeypatch
):
    """RSDL_SHM_QUEUE=1: batch tensors cross the named-queue socket as
    shared-memory handles (torch ForkingPickler, file_system strategy)
    instead of pickled bytes. Same correctness contract as the plain
    transport test."""
    monkeypatch.setenv("RSDL_SHM_QUEUE", "1")
    from ray_shuffling_data_loader_amd.data_generation import generate_data
    from ray_shuffling_data_loader_amd.dataset import ShufflingDataset

    num_rows = 6000
    filenames, _ = generate_data(num_rows, 2, 1, 0.0, str(tmp_path))
    filenames = list(filenames)
    qname = f"mp_shm_{time.time_ns()}"

    ctx = mp_spawn_context
    result_q = ctx.Queue()
    p = ctx.Process(
        target=_rank1_consumer,
        args=(qname, filenames, num_rows, result_q),
    )
    ds0 = ShufflingDataset(
        filenames,
        num_epochs=1,
        num_trainers=2,
        batch_size=1000,
        rank=0,
        num_reducers=4,
        queue_name=qname,
        seed=5,
    )
    p.start()
    ds0.set_epoch(0)
    keys0 = torch.cat([b["key"] for b in ds0]).tolist()
    out = result_q.get(timeout=120)
    assert not isinstance(out, str), out
    p.join(timeout=60)
    assert p.exitcode == 0
    assert sorted(keys0 + out) == list(range(num_rows))


def _shm_get_one(qname, result_q):
    try:
        import os

        os.environ["RSDL_SHM_QUEUE"] = "1"
        from ray_shuffling_data_loader_amd.batch_queue import BatchQueue

        q = BatchQueue(1, 1, 1, name=qname, connect=True)
        t = q.get(0, 0, timeout=30)
        # Prove the tensor crossed as a shared-memory handle, not bytes.
        result_q.put((t.is_shared(), t.tolist()))
    except Exception as e:
        import traceback

        result_q.put(f"ERROR: {e}\n{traceback.format_exc()}")


def test_shm_transport_is_zero_copy(mp_spawn_context, monkeypatch):
    monkeypatch.setenv("RSDL_SHM_QUEUE", "1")
    from ray_shuffling_data_loader_amd.batch_queue import BatchQueue

    qname = f"shm_zc_{time.time_ns()}"
    q = BatchQueue(1, 1, 1, name=qname)
    q.new_epoch(0)
    q.put(0, 0, torch.arange(16, dtype=torch.float32))
    ctx = mp_spawn_context
    result_q = ctx.Queue()
    p = ctx.Process(target=_shm_get_one, args=(qname, result_q))
    p.start()
    out = result_q.get(timeout=60)
    p.join(timeout=30)
    assert not isinstance(out, str), out
    is_shared, values = out
    assert is_shared, "tensor should arrive shm-backed under RSDL_SHM_QUEUE"
    assert values == list(range(16))
    q.shutdown()
