"""Shuffle driver API.

Parity with the reference's functional surface (reference:
ray_shuffling_data_loader/shuffle.py:11-219): the ``BatchConsumer`` interface
and a ``shuffle(...)`` entry point that drives ``num_epochs`` shuffles of the
dataset into the consumer under the epoch-window backpressure gate. The
actual pipeline lives in :class:`ray_shuffling_data_loader_amd.engine.
ShuffleEngine` (map -> RCCL all-to-all -> fused permute on MI355X).
"""

from typing import List, Sequence

from ray_shuffling_data_loader_amd.engine import ShuffleEngine


class BatchConsumer:
    """Interface for consumers of shuffle outputs
    (reference shuffle.py:11-43)."""

    def consume(self, rank, epoch, batches):
        """Consume the provided batches for the given trainer and epoch."""
        raise NotImplementedError(
            "Derived classes must implement consume method."
        )

    def producer_done(self, rank, epoch):
        """Signals that production is done for the given trainer/epoch."""
        raise NotImplementedError(
            "Derived classes must implement producer_done method."
        )

    def wait_until_ready(self, epoch):
        """Returns once the consumer is ready for this epoch to start."""
        raise NotImplementedError(
            "Derived classes must implement wait_until_ready method."
        )

    def wait_until_all_epochs_done(self):
        """Returns once all batches for all epochs have been consumed."""
        raise NotImplementedError(
            "Derived classes must implement wait_until_done method."
        )


def shuffle(
    filenames: Sequence[str],
    batch_consumer: BatchConsumer,
    num_epochs: int,
    num_reducers: int,
    num_trainers: int,
    stats_collector=None,
    **engine_kwargs,
) -> float:
    """Shuffle the dataset into ``batch_consumer`` every epoch; returns the
    wall duration (reference shuffle.py:51-86). Synchronous — run it in a
    thread (or use ShuffleEngine.start()) for pipelined operation."""
    engine = ShuffleEngine(
        filenames,
        batch_consumer,
        num_epochs=num_epochs,
        num_reducers=num_reducers,
        num_trainers=num_trainers,
        stats_collector=stats_collector,
        **engine_kwargs,
    )
    return engine.run()


# Small LRU of engines for repeated shuffle_epoch calls: each cached
# engine may pin an HBM-resident source block, so the cache is bounded.
_EPOCH_ENGINES = {}
_EPOCH_ENGINES_MAX = 4


def shuffle_epoch(
    epoch: int,
    filenames: Sequence[str],
    batch_consumer: BatchConsumer,
    num_reducers: int,
    num_trainers: int,
    stats_collector=None,
    engine: ShuffleEngine = None,
    **engine_kwargs,
) -> None:
    """Shuffle one epoch into the consumer (reference shuffle.py:89-126).

    Calls with the same (filenames, reducers, trainers) reuse one cached
    engine, so the source stays resident (HBM/host per ``source_cache``)
    instead of being re-read and re-packed per invocation — calling this in
    a loop costs one ingest total, like :class:`ShuffleEngine` itself.
    Pass ``engine=`` to manage the instance explicitly.
    """
    if engine is None:
        key = (tuple(filenames), num_reducers, num_trainers)
        engine = _EPOCH_ENGINES.get(key)
        if engine is None:
            engine = ShuffleEngine(
                filenames,
                batch_consumer,
                num_epochs=epoch + 1,
                num_reducers=num_reducers,
                num_trainers=num_trainers,
                stats_collector=stats_collector,
                **engine_kwargs,
            )
            while len(_EPOCH_ENGINES) >= _EPOCH_ENGINES_MAX:
                _EPOCH_ENGINES.pop(next(iter(_EPOCH_ENGINES)))
            _EPOCH_ENGINES[key] = engine
        else:
            # LRU refresh
            _EPOCH_ENGINES.pop(key, None)
            _EPOCH_ENGINES[key] = engine
        engine.num_epochs = max(engine.num_epochs, epoch + 1)
        engine.consumer = batch_consumer
        engine.stats = stats_collector
    engine._shuffle_epoch(epoch)


def consume(
    rank: int, batch_consumer: BatchConsumer, epoch: int, batches: List
) -> None:
    """Deliver batches to the consumer and signal producer-done
    (reference shuffle.py:203-219)."""
    batch_consumer.consume(rank, epoch, batches)
    batch_consumer.producer_done(rank, epoch)
