"""Property-based tests (hypothesis) for the leftover-carry rebatch loop —
the logic SURVEY.md §7 flags as deadlock/off-by-one prone (and where the
reference silently drops tails, reference dataset.py:160-168)."""

import torch
from hypothesis import given, settings, strategies as st

from ray_shuffling_data_loader_amd.batch_queue import BatchQueue
from ray_shuffling_data_loader_amd.dataset import ShufflingDataset
from ray_shuffling_data_loader_amd.utils.rowblock import RowBlock


def _run_rebatch(block_sizes, batch_size, drop_last, return_queue=False):
    """Drive ShufflingDataset.__iter__ directly over a hand-built queue."""
    total = sum(block_sizes)
    q = BatchQueue(2, 1, 1)
    q.new_epoch(0)
    start = 0
    for n in block_sizes:
        q.put(0, 0, RowBlock({"v": torch.arange(start, start + n)}))
        start += n
    q.producer_done(0, 0)

    ds = ShufflingDataset.__new__(ShufflingDataset)
    ds._batch_queue = q
    ds._qrank = 0
    ds._epoch = 0
    ds._last_epoch = None
    ds._batch_size = batch_size
    ds._drop_last = drop_last
    ds._num_epochs = 1
    ds._engine = None
    out = list(iter(ds))
    if return_queue:
        return total, out, q
    return total, out


@settings(max_examples=200, deadline=None)
@given(
    block_sizes=st.lists(st.integers(0, 70), min_size=0, max_size=20),
    batch_size=st.integers(1, 97),
    drop_last=st.booleans(),
)
def test_rebatch_invariants(block_sizes, batch_size, drop_last):
    total, out = _run_rebatch(block_sizes, batch_size, drop_last)
    tail = total % batch_size
    n_full = total // batch_size
    # 1. every batch except a possible final tail is exactly batch_size
    for b in out[: n_full]:
        assert len(b) == batch_size
    # 2. row count: tail kept iff not drop_last (the reference drops some
    #    tails silently; ours must never lose a row unless asked)
    expect = total - (tail if drop_last else 0)
    assert sum(len(b) for b in out) == expect
    if not drop_last and tail:
        assert len(out[-1]) == tail
    # 3. order-preserving: concatenation reproduces the input stream
    if out:
        cat = torch.cat([b["v"] for b in out])
        assert torch.equal(cat, torch.arange(expect))


@settings(max_examples=50, deadline=None)
@given(
    block_sizes=st.lists(st.integers(0, 40), min_size=1, max_size=10),
    batch_size=st.integers(1, 50),
)
def test_rebatch_epoch_accounting(block_sizes, batch_size):
    """After a full consume (incl. the sentinel task_done), the epoch's
    queues are joined: with max_concurrent_epochs=1 the next epoch window
    must open without blocking (the §7(a) deadlock hazard)."""
    import threading

    total, out, q = _run_rebatch(block_sizes, batch_size, False, True)
    assert sum(len(b) for b in out) == total
    opened = threading.Event()

    def open_next():
        q.new_epoch(1)
        opened.set()

    t = threading.Thread(target=open_next, daemon=True)
    t.start()
    t.join(timeout=15)
    assert opened.is_set(), (
        "epoch 1 window did not open: join/task_done accounting is off"
    )
