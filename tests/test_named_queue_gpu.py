"""GPU test: named-socket queue mode with the shared-memory/dmabuf
transport (default RSDL_SHM_QUEUE=1).

Two trainer processes on one GPU: rank 0 owns the queue + engine (batches
land HBM-resident), rank 1 connects over the Unix socket and consumes its
partition — tensor payloads cross as device-IPC (dmabuf) handles, not
pickled bytes. The analog of the reference's plasma zero-copy consumer get
(reference dataset.py:136-139).
"""

import os
import sys
import tempfile
import time

import pytest

pytestmark = pytest.mark.gpu


def _rank1_consume(qname, filenames, result_q):
    try:
        sys.path.insert(
            0,
            os.path.dirname(os.path.dirname(os.path.abspath(__file__))),
        )
        from ray_shuffling_data_loader_amd.dataset import ShufflingDataset

        ds = ShufflingDataset(
            filenames,
            1,
            num_trainers=2,
            batch_size=20_000,
            rank=1,
            num_reducers=4,
            queue_name=qname,
        )
        ds.set_epoch(0)
        total = 0
        cuda_batches = 0
        checksum = 0.0
        for b in ds:
            total += len(b)
            cuda_batches += int(b.device.type == "cuda")
            # Touch the payload: proves the IPC mapping is readable.
            checksum += float(
                next(iter(b.columns.values()))[:16].float().sum().item()
            )
        result_q.put(("ok", total, cuda_batches, checksum))
    except Exception as e:  # noqa: BLE001 - surfaced in parent assert
        import traceback

        result_q.put(("err", str(e), traceback.format_exc()))


def test_named_queue_shm_transport_gpu(mp_spawn_context):
    import torch

    from ray_shuffling_data_loader_amd.data_generation import (
        float_data_spec,
        generate_data,
    )
    from ray_shuffling_data_loader_amd.dataset import ShufflingDataset

    assert os.environ.get("RSDL_SHM_QUEUE", "1") == "1", (
        "this test validates the default shm transport"
    )
    num_rows = 200_000
    d = tempfile.mkdtemp()
    filenames, _ = generate_data(
        num_rows, 2, 1, 0.0, d, spec=float_data_spec(16), include_key=False
    )
    filenames = list(filenames)
    qname = f"gpu_shm_{time.time_ns()}"
    rq = mp_spawn_context.Queue()
    p = mp_spawn_context.Process(
        target=_rank1_consume, args=(qname, filenames, rq)
    )
    ds0 = ShufflingDataset(
        filenames,
        1,
        num_trainers=2,
        batch_size=20_000,
        rank=0,
        num_reducers=4,
        queue_name=qname,
        device=torch.device("cuda", 0),
    )
    p.start()
    try:
        ds0.set_epoch(0)
        rank0_rows = sum(len(b) for b in ds0)
        status = rq.get(timeout=180)
        p.join(timeout=60)
        assert status[0] == "ok", status
        _, rank1_rows, cuda_batches, checksum = status
        assert rank0_rows + rank1_rows == num_rows
        # The engine produced on GPU; rank 1's batches must have arrived as
        # device tensors (IPC), not host copies of pickled bytes.
        assert cuda_batches > 0, "rank 1 saw no CUDA batches over the socket"
        assert checksum == checksum  # not NaN: payload readable
    finally:
        if p.is_alive():
            p.terminate()
            p.join(timeout=30)
