"""Direct tests of the functional shuffle driver surface
(reference shuffle.py:51-126, 203-219): shuffle() / shuffle_epoch() /
consume() against a custom BatchConsumer."""

import threading

import torch

from ray_shuffling_data_loader_amd.data_generation import generate_data
from ray_shuffling_data_loader_amd.shuffle import (
    BatchConsumer,
    consume,
    shuffle,
    shuffle_epoch,
)


class CollectingConsumer(BatchConsumer):
    def __init__(self, num_trainers, num_epochs):
        self.rows = {
            (e, r): []
            for e in range(num_epochs)
            for r in range(num_trainers)
        }
        self.done = {k: False for k in self.rows}
        self._lock = threading.Lock()

    def consume(self, rank, epoch, batches):
        with self._lock:
            for b in batches:
                self.rows[(epoch, rank)].append(b["key"])

    def producer_done(self, rank, epoch):
        self.done[(epoch, rank)] = True

    def wait_until_ready(self, epoch):
        return

    def wait_until_all_epochs_done(self):
        return


def test_shuffle_driver_delivers_every_row(tmp_path):
    num_rows, trainers, epochs = 5000, 2, 3
    filenames, _ = generate_data(num_rows, 2, 1, 0.0, str(tmp_path))
    c = CollectingConsumer(trainers, epochs)
    duration = shuffle(
        list(filenames), c, epochs, num_reducers=4, num_trainers=trainers,
        seed=3,
    )
    assert duration > 0
    assert all(c.done.values())
    orders = []
    for e in range(epochs):
        keys = torch.cat(
            [t for r in range(trainers) for t in c.rows[(e, r)]]
        )
        assert sorted(keys.tolist()) == list(range(num_rows)), e
        orders.append(keys.tolist())
    # epochs are reshuffled (astronomically unlikely to repeat)
    assert orders[0] != orders[1]


def test_shuffle_epoch_single(tmp_path):
    num_rows = 3000
    filenames, _ = generate_data(num_rows, 2, 1, 0.0, str(tmp_path))
    c = CollectingConsumer(1, 1)
    shuffle_epoch(
        0, list(filenames), c, num_reducers=2, num_trainers=1
    )
    keys = torch.cat(c.rows[(0, 0)])
    assert sorted(keys.tolist()) == list(range(num_rows))


def test_consume_helper():
    c = CollectingConsumer(1, 1)
    from ray_shuffling_data_loader_amd.utils.rowblock import RowBlock

    consume(0, c, 0, [RowBlock({"key": torch.arange(5)})])
    assert c.done[(0, 0)]
    assert torch.equal(c.rows[(0, 0)][0], torch.arange(5))


def test_shuffle_epoch_reuses_cached_engine(tmp_path):
    """Repeated shuffle_epoch calls with the same (files, reducers,
    trainers) must reuse ONE engine (one ingest total) instead of
    re-reading the source per call, and the cache stays bounded."""
    import importlib

    sh = importlib.import_module("ray_shuffling_data_loader_amd.shuffle")

    sh._EPOCH_ENGINES.clear()
    filenames, _ = generate_data(1000, 2, 1, 0.0, str(tmp_path))
    c = CollectingConsumer(1, 3)
    for e in range(3):
        shuffle_epoch(e, list(filenames), c, num_reducers=2,
                      num_trainers=1)
    assert len(sh._EPOCH_ENGINES) == 1
    eng = next(iter(sh._EPOCH_ENGINES.values()))
    assert eng.num_epochs >= 3
    for e in range(3):
        keys = torch.cat(c.rows[(e, 0)])
        assert sorted(keys.tolist()) == list(range(1000)), e

    # Distinct configs evict LRU-style at the bound (4).
    for r in (3, 4, 5, 6):
        c2 = CollectingConsumer(1, 1)
        shuffle_epoch(0, list(filenames), c2, num_reducers=r,
                      num_trainers=1)
    assert len(sh._EPOCH_ENGINES) == sh._EPOCH_ENGINES_MAX
    # the original (reducers=2) entry was the oldest -> evicted
    assert all(k[1] != 2 for k in sh._EPOCH_ENGINES)
    sh._EPOCH_ENGINES.clear()


def test_shuffle_epoch_explicit_engine(tmp_path):
    """Passing engine= bypasses the cache entirely."""
    import importlib

    sh = importlib.import_module("ray_shuffling_data_loader_amd.shuffle")
    from ray_shuffling_data_loader_amd.engine import ShuffleEngine

    sh._EPOCH_ENGINES.clear()
    filenames, _ = generate_data(500, 1, 1, 0.0, str(tmp_path))
    c = CollectingConsumer(1, 1)
    eng = ShuffleEngine(list(filenames), c, num_epochs=1,
                        num_reducers=2, num_trainers=1)
    shuffle_epoch(0, list(filenames), c, num_reducers=2,
                  num_trainers=1, engine=eng)
    assert len(sh._EPOCH_ENGINES) == 0
    keys = torch.cat(c.rows[(0, 0)])
    assert sorted(keys.tolist()) == list(range(500))
    sh._EPOCH_ENGINES.clear()
