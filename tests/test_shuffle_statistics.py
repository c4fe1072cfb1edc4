"""Statistical contract tests: the engine's one-permutation+multinomial
construction must match the reference's two-stage randomness (iid uniform
reducer assignment + per-reducer permutation) in distribution. CPU, seeded.
"""

import numpy as np
import pytest
import torch

from ray_shuffling_data_loader_amd.data_generation import generate_data
from ray_shuffling_data_loader_amd.engine import ShuffleEngine
from ray_shuffling_data_loader_amd.shuffle import BatchConsumer


class CollectSink(BatchConsumer):
    def __init__(self):
        self.partitions = {}  # (rank, epoch) -> list of RowBlocks

    def consume(self, rank, epoch, batches):
        self.partitions.setdefault((rank, epoch), []).extend(batches)

    def producer_done(self, rank, epoch):
        pass

    def wait_until_ready(self, epoch):
        pass

    def wait_until_all_epochs_done(self):
        pass


@pytest.fixture(scope="module")
def data(tmp_path_factory):
    d = tmp_path_factory.mktemp("stat_parquet")
    num_rows = 40000
    filenames, _ = generate_data(num_rows, 2, 1, 0.0, str(d))
    return list(filenames), num_rows


def run_epochs(filenames, num_rows, num_reducers, epochs, seed):
    sink = CollectSink()
    eng = ShuffleEngine(
        filenames,
        sink,
        num_epochs=epochs,
        num_reducers=num_reducers,
        num_trainers=1,
        seed=seed,
        device=torch.device("cpu"),
    )
    eng.run()
    return sink


def test_partition_sizes_look_multinomial(data):
    filenames, num_rows = data
    R = 8
    sizes = []
    for seed in range(5):
        sink = run_epochs(filenames, num_rows, R, 1, seed=100 + seed)
        parts = sink.partitions[(0, 0)]
        assert len(parts) == R
        sizes.extend(len(p) for p in parts)
    sizes = np.array(sizes, dtype=np.float64)
    mean = num_rows / R
    # Binomial(num_rows, 1/R): std = sqrt(n p (1-p)) ~ 66 for 40000/8.
    std = np.sqrt(num_rows * (1 / R) * (1 - 1 / R))
    assert abs(sizes.mean() - mean) < 3 * std / np.sqrt(len(sizes))
    # Sizes must VARY (binomial, not an exact equal split).
    assert sizes.std() > 0.3 * std
    assert sizes.std() < 3.0 * std


def test_within_partition_order_uniform(data):
    # The average normalized position of each key across many epochs should
    # concentrate around 0.5 (uniform placement), and positions should
    # decorrelate across epochs.
    filenames, num_rows = data
    epochs = 6
    sink = run_epochs(filenames, num_rows, 4, epochs, seed=7)
    pos = np.zeros(num_rows)
    for e in range(epochs):
        parts = sink.partitions[(0, e)]
        keys = torch.cat([p["key"] for p in parts]).numpy()
        order = np.empty(num_rows)
        order[keys] = np.arange(num_rows) / num_rows
        pos += order
    pos /= epochs
    # Mean position per row ~ Uniform(0,1) averaged over 6 epochs:
    # std = 1/sqrt(12*6) ~ 0.118.
    assert abs(pos.mean() - 0.5) < 0.01
    assert 0.08 < pos.std() < 0.16
    # No strong correlation with original key order (would indicate a
    # non-uniform shuffle).
    corr = np.corrcoef(np.arange(num_rows), pos)[0, 1]
    assert abs(corr) < 0.05


def test_uneven_reducer_split_weights():
    # num_reducers not divisible by num_trainers: destination-trainer
    # weights must be proportional to owned reducers (array_split parity).
    eng = ShuffleEngine.__new__(ShuffleEngine)
    splits = np.array_split(np.arange(7), 3)
    assert [len(s) for s in splits] == [3, 2, 2]
