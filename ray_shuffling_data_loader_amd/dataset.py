"""ShufflingDataset: framework-agnostic per-epoch shuffling dataset.

API parity with the reference (reference: ray_shuffling_data_loader/
dataset.py:15-205): same constructor signature, ``set_epoch`` guard with the
same error, the same leftover-carry re-batching loop, ``drop_last``, the
``num_reducers = num_trainers x cores x 0.6`` default and
``max_concurrent_epochs=2``. Batches are :class:`RowBlock`s (GPU-resident
column tensors) instead of pandas DataFrames; ``RowBlock.to_pandas()`` gives
the old representation when needed.

Process modes (replacing the reference's rank-0 queue actor + shuffle task,
dataset.py:52-84):

  * **distributed** (torch.distributed initialized, world > 1): every rank
    runs a symmetric ShuffleEngine worker thread; the map->reduce exchange is
    an RCCL all-to-all over xGMI and each rank consumes its own in-process
    queue. This is the MI355X flagship path — data never crosses a process
    boundary outside the collective.
  * **local** (no distributed init): rank 0 owns the queue and an engine
    producing for all trainers; ranks > 0 connect to rank 0's named queue
    over a Unix socket (parity with the named-actor connect pattern).
"""

import os

from typing import List, Optional

import torch

from ray_shuffling_data_loader_amd.batch_queue import BatchQueue
from ray_shuffling_data_loader_amd.engine import (
    ShuffleEngine,
    ShuffleEngineFailure,
)
from ray_shuffling_data_loader_amd.shuffle import BatchConsumer
from ray_shuffling_data_loader_amd.utils.rowblock import RowBlock
from ray_shuffling_data_loader_amd.parallel import fabric

BATCHQUEUE_ACTOR_NAME = "BatchQueue"
REDUCER_CLUSTER_CORE_SHARE = 0.6


class BatchConsumerQueue(BatchConsumer):
    """Adapter: BatchConsumer interface -> BatchQueue ops (reference
    dataset.py:191-205)."""

    def __init__(self, batch_queue: BatchQueue, rank_map=None):
        self._batch_queue = batch_queue
        # In the symmetric distributed mode each rank's queue is local and
        # single-column; global trainer ids map onto column 0.
        self._rank_map = rank_map or {}

    def _q(self, rank: int) -> int:
        return self._rank_map.get(rank, rank)

    def consume(self, rank: int, epoch: int, batches: List):
        self._batch_queue.put_batch(self._q(rank), epoch, batches)

    def producer_done(self, rank: int, epoch: int):
        self._batch_queue.producer_done(self._q(rank), epoch)

    def wait_until_ready(self, epoch: int):
        self._batch_queue.new_epoch(epoch)

    def wait_until_all_epochs_done(self):
        self._batch_queue.wait_until_all_epochs_done()


class ShufflingDataset:
    """A shuffling dataset that yields batches upon iteration.

    Shuffling for up to ``max_concurrent_epochs`` epochs is kicked off at
    construction time (reference dataset.py:15-94).

    Args:
        filenames: Paths to input Parquet files.
        num_epochs: Number of training epochs.
        num_trainers: Number of trainer workers.
        batch_size: Size of the batches the iterator yields.
        rank: The worker rank of the current process.
        drop_last: Whether to drop the last incomplete batch. Default False.
        num_reducers: Number of shuffler reducers. Default
            num_trainers x cores x 0.6 (reference dataset.py:46-48).
        max_concurrent_epochs: Max epochs shuffling concurrently. Default 2.
        engine_kwargs: MI355X engine options (device, source_cache,
            feature_matrix, seed, ...). See ShuffleEngine.
    """

    def __init__(
        self,
        filenames: List[str],
        num_epochs: int,
        num_trainers: int,
        batch_size: int,
        rank: int,
        drop_last: bool = False,
        num_reducers: Optional[int] = None,
        max_concurrent_epochs: int = 2,
        queue_name: str = BATCHQUEUE_ACTOR_NAME,
        start_epoch: int = 0,
        **engine_kwargs,
    ):
        if num_reducers is None:
            num_reducers = int(
                num_trainers
                * (os.cpu_count() or 1)
                * REDUCER_CLUSTER_CORE_SHARE
            )
            num_reducers = max(num_reducers, num_trainers)

        self._batch_size = batch_size
        self._num_epochs = num_epochs
        self._num_trainers = num_trainers
        self._rank = rank
        self._drop_last = drop_last
        self._epoch = None
        self._last_epoch = None
        self._engine = None

        world, _, initialized = fabric.dist_info()
        distributed = initialized and (
            world > 1 or os.environ.get("RSDL_FORCE_COLLECTIVE") == "1"
        )

        if distributed:
            # Symmetric mode: each rank owns a local single-column queue
            # (its own partition stream); the epoch window gates locally and
            # ranks align at the exchange collective.
            self._qrank = 0
            self._batch_queue = BatchQueue(
                num_epochs, 1, max_concurrent_epochs
            )
            consumer = BatchConsumerQueue(
                self._batch_queue, rank_map={rank: 0}
            )
            self._engine = ShuffleEngine(
                filenames,
                consumer,
                num_epochs=num_epochs,
                num_reducers=num_reducers,
                num_trainers=num_trainers,
                rank=rank,
                start_epoch=start_epoch,
                **engine_kwargs,
            )
            self._engine.start()
        elif rank == 0:
            self._qrank = rank
            # Local central mode (reference dataset.py:52-74): rank 0 creates
            # the queue (served over a named socket when other trainer
            # processes will connect) and kicks off the shuffle driver.
            self._batch_queue = BatchQueue(
                num_epochs,
                num_trainers,
                max_concurrent_epochs,
                name=queue_name if num_trainers > 1 else None,
                connect=False,
            )
            consumer = BatchConsumerQueue(self._batch_queue)
            self._batch_queue.ready()
            self._engine = ShuffleEngine(
                filenames,
                consumer,
                num_epochs=num_epochs,
                num_reducers=num_reducers,
                num_trainers=num_trainers,
                rank=0,
                start_epoch=start_epoch,
                **engine_kwargs,
            )
            self._engine.start()
        else:
            self._qrank = rank
            # Worker process: connect to rank 0's queue
            # (reference dataset.py:76-84).
            self._batch_queue = BatchQueue(
                num_epochs,
                num_trainers,
                max_concurrent_epochs,
                name=queue_name,
                connect=True,
            )

    def set_epoch(self, epoch):
        """Set the current training epoch; call before constructing the
        iterator each epoch (reference dataset.py:96-106)."""
        self._epoch = epoch

    def __iter__(self):
        """Yields RowBlock batches of ``batch_size`` rows from the shuffle
        queue (reference dataset.py:108-188)."""
        if self._epoch is None or self._epoch == self._last_epoch:
            raise ValueError(
                "You must set the epoch on this dataset via set_epoch()"
                "at the beginning of each epoch, before iterating over this "
                "dataset (e.g. via enumerate(ds))."
            )

        buffer = None  # leftover-carry block
        is_done = False
        while not is_done:
            pending = self._batch_queue.get_batch(self._qrank, self._epoch)
            if pending and pending[-1] is None:
                is_done = True
                pending.pop()
            num_outstanding = len(pending)

            for block in pending:
                if isinstance(block, ShuffleEngineFailure):
                    raise RuntimeError(
                        "shuffle engine worker failed"
                    ) from block.error
                if len(block) == 0:
                    continue
                if block.device.type == "cuda":
                    # Engine partitions are allocated on the side shuffle
                    # stream; bind their lifetime to the consumer's stream
                    # so dropping a batch mid-kernel can't recycle its
                    # memory into the pipelined next-epoch shuffle.
                    block.record_stream(
                        torch.cuda.current_stream(block.device)
                    )
                buffer_len = len(buffer) if buffer is not None else 0
                offset = self._batch_size - buffer_len
                buffer = RowBlock.concat([buffer, block.slice(0, offset)])
                if len(buffer) == self._batch_size:
                    yield buffer
                    buffer = None
                # Full batches from the rest of the block, then save the
                # tail. NOTE: diverges from the reference's pos bookkeeping
                # (dataset.py:160-168), which silently DROPS the tail when
                # 0 < len(block) - offset < batch_size (its `pos` fallback
                # overshoots); here the tail is always carried.
                n_rest = len(block) - offset
                n_full = n_rest // self._batch_size if n_rest > 0 else 0
                for k in range(n_full):
                    start = offset + k * self._batch_size
                    yield block.slice(start, start + self._batch_size)
                tail = offset + n_full * self._batch_size
                if 0 < len(block) - tail:
                    buffer = block.slice(tail)

            if num_outstanding > 0:
                self._batch_queue.task_done(
                    self._qrank, self._epoch, num_outstanding
                )

        if buffer is not None and not self._drop_last:
            yield buffer
        # Acknowledge the producer-done sentinel.
        self._batch_queue.task_done(self._qrank, self._epoch, 1)
        self._last_epoch = self._epoch
        if self._epoch == self._num_epochs - 1 and self._engine is not None:
            self._engine.join()


if __name__ == "__main__":
    import shutil
    import tempfile

    from ray_shuffling_data_loader_amd.data_generation import generate_data
    from ray_shuffling_data_loader_amd.utils.stats import human_readable_size

    num_rows = 10**6
    num_files = 10
    data_dir = tempfile.mkdtemp()
    print(
        f"Generating {num_rows} rows over {num_files} files, with 1 row "
        "group per file."
    )
    filenames, num_bytes = generate_data(
        num_rows, num_files, 1, 0.0, data_dir
    )
    print(
        f"Generated {len(filenames)} files containing {num_rows} rows "
        f"totalling {human_readable_size(num_bytes)}."
    )
    num_epochs = 4
    batch_size = 20000
    num_reducers = 8
    print(
        f"Creating shuffling dataset with {batch_size} batch size, "
        f"{num_epochs} epochs, {num_reducers} reducers, and 1 trainer."
    )
    print(f"Should consume {num_rows // batch_size} batches.")
    ds = ShufflingDataset(
        list(filenames),
        num_epochs,
        1,
        batch_size,
        0,
        num_reducers=num_reducers,
    )
    for epoch in range(num_epochs):
        ds.set_epoch(epoch)
        for batch_idx, batch in enumerate(ds):
            print(f"Consuming batch {batch_idx}!")
    print("Done consuming batches.")
    shutil.rmtree(data_dir)
