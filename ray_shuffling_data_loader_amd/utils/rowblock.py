"""RowBlock: the unit of data flowing through the shuffle pipeline.

The MI355X-native analog of the pandas DataFrames the reference passes
through the Ray object store (reference: ray_shuffling_data_loader/
shuffle.py:151-199, dataset.py:139-168): a set of equal-length torch column
tensors that may live on GPU (HBM-resident shuffle output) or CPU. Slicing
returns views (zero-copy — the re-batching loop in ShufflingDataset.__iter__
slices thousands of these); concat copies, like ``pd.concat``.
"""

from typing import Dict, List, Optional

import torch


class RowBlock:
    """Equal-length named column tensors. Column tensors are 1-D
    ``[num_rows]`` or 2-D ``[num_rows, numel]``."""

    __slots__ = ("columns", "_len")

    def __init__(self, columns: Dict[str, torch.Tensor]):
        if not columns:
            raise ValueError("RowBlock needs at least one column")
        n = None
        for name, t in columns.items():
            if n is None:
                n = t.shape[0]
            elif t.shape[0] != n:
                raise ValueError(
                    f"column {name} has {t.shape[0]} rows, expected {n}"
                )
        self.columns = columns
        self._len = int(n)

    def __len__(self) -> int:
        return self._len

    @property
    def device(self) -> torch.device:
        return next(iter(self.columns.values())).device

    @property
    def names(self) -> List[str]:
        return list(self.columns.keys())

    def __getitem__(self, key):
        if isinstance(key, str):
            return self.columns[key]
        if isinstance(key, slice):
            return self.slice(key.start or 0, key.stop)
        raise TypeError(f"unsupported index type: {type(key)}")

    def slice(self, start: int, stop: Optional[int] = None) -> "RowBlock":
        """Zero-copy row range view."""
        stop = self._len if stop is None else min(stop, self._len)
        return RowBlock({k: t[start:stop] for k, t in self.columns.items()})

    @staticmethod
    def concat(blocks: List["RowBlock"]) -> "RowBlock":
        blocks = [b for b in blocks if b is not None and len(b) > 0]
        if not blocks:
            raise ValueError("nothing to concat")
        if len(blocks) == 1:
            return blocks[0]
        names = blocks[0].names
        return RowBlock(
            {k: torch.cat([b.columns[k] for b in blocks]) for k in names}
        )

    def to(self, device, non_blocking: bool = False) -> "RowBlock":
        return RowBlock(
            {
                k: t.to(device, non_blocking=non_blocking)
                for k, t in self.columns.items()
            }
        )

    def pin_memory(self) -> "RowBlock":
        return RowBlock({k: t.pin_memory() for k, t in self.columns.items()})

    def record_stream(self, stream: "torch.cuda.Stream") -> None:
        """Mark every CUDA column as in use on ``stream``. The shuffle engine
        allocates partitions on its side HIP stream; without this, dropping a
        batch whose consumer-stream kernels are still in flight lets the
        caching allocator hand the block back to the side-stream pool, where
        the pipelined next-epoch shuffle (max_concurrent_epochs=2) could
        overwrite it mid-read. Columns often share one storage (views mode);
        record_stream is per-storage underneath, so duplicates are cheap."""
        seen = set()
        for t in self.columns.values():
            if t.device.type != "cuda":
                continue
            key = t.untyped_storage().data_ptr()
            if key in seen:
                continue
            seen.add(key)
            t.record_stream(stream)

    def contiguous(self) -> "RowBlock":
        return RowBlock({k: t.contiguous() for k, t in self.columns.items()})

    # ----- interop -----------------------------------------------------------

    def to_pandas(self):
        """Adapter for code written against the reference's DataFrame
        batches."""
        import pandas as pd

        out = {}
        for k, t in self.columns.items():
            a = t.detach().cpu().numpy()
            if a.ndim == 2 and a.shape[1] == 1:
                a = a[:, 0]
            if a.ndim == 2:
                out[k] = list(a)
            else:
                out[k] = a
        return pd.DataFrame(out)

    @staticmethod
    def from_pandas(df) -> "RowBlock":
        import numpy as np

        cols = {}
        for k in df.columns:
            v = df[k].values
            if v.dtype == object:
                v = np.stack(v)
            cols[k] = torch.from_numpy(np.ascontiguousarray(v))
        return RowBlock(cols)

    @staticmethod
    def from_numpy(arrays) -> "RowBlock":
        return RowBlock(
            {k: torch.from_numpy(v) for k, v in arrays.items()}
        )

    def __repr__(self) -> str:
        cols = ", ".join(
            f"{k}:{tuple(t.shape)}:{t.dtype}" for k, t in self.columns.items()
        )
        return f"RowBlock(len={self._len}, device={self.device}, [{cols}])"
