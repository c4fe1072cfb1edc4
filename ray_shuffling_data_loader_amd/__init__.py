"""MI355X-native per-epoch shuffling data loader.

A from-scratch re-design of ray-project/ray_shuffling_data_loader for AMD
Instinct MI355X (gfx950): PyTorch-ROCm + hand-written HIP/CDNA4 kernels for
the fused shuffle ops + RCCL all-to-all over xGMI for the map->reduce
exchange. Same public surface as the reference
(reference: ray_shuffling_data_loader/__init__.py:1-7).
"""

from ray_shuffling_data_loader_amd.dataset import ShufflingDataset
from ray_shuffling_data_loader_amd.shuffle import BatchConsumer, shuffle

__version__ = "0.1.0"

__all__ = [
    "ShufflingDataset",
    "TorchShufflingDataset",
    "BatchConsumer",
    "shuffle",
    "__version__",
]


def __getattr__(name):
    # torch_dataset pulls in torch; keep it lazy so the queue/data layers
    # stay importable in torch-less tooling contexts.
    if name == "TorchShufflingDataset":
        from ray_shuffling_data_loader_amd.torch_dataset import (
            TorchShufflingDataset,
        )

        return TorchShufflingDataset
    raise AttributeError(name)
