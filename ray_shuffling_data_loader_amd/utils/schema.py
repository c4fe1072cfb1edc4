"""Column schema + packed-row layout for the shuffle exchange.

The reference moves pandas DataFrames through the Ray object store; here the
unit crossing the wire (RCCL all-to-all over xGMI) is a packed row-major byte
matrix, described by a :class:`Schema`. Columns are laid out inside each row
in descending dtype-size order so every column offset is naturally aligned,
and the row stride is padded to 16 B so vectorized (dwordx4) kernels stay
aligned across rows.

Reference analog: the implicit schema of ``DATA_SPEC``
(reference: ray_shuffling_data_loader/data_generation.py:56-77) and the
per-column dtype handling of ``convert_to_tensor``
(reference: ray_shuffling_data_loader/torch_dataset.py:204-236).
"""

from dataclasses import dataclass, field
from typing import Dict, List, Optional

import numpy as np
import torch

_DTYPE_BYTES = {
    torch.float64: 8,
    torch.int64: 8,
    torch.float32: 4,
    torch.int32: 4,
    torch.float16: 2,
    torch.bfloat16: 2,
    torch.int16: 2,
    torch.uint8: 1,
    torch.int8: 1,
    torch.bool: 1,
}

NUMPY_TO_TORCH_DTYPE = {
    np.dtype(np.bool_): torch.bool,
    np.dtype(np.uint8): torch.uint8,
    np.dtype(np.int8): torch.int8,
    np.dtype(np.int16): torch.int16,
    np.dtype(np.int32): torch.int32,
    np.dtype(np.int64): torch.int64,
    np.dtype(np.float16): torch.float16,
    np.dtype(np.float32): torch.float32,
    np.dtype(np.float64): torch.float64,
}

TORCH_TO_NUMPY_DTYPE = {v: k for k, v in NUMPY_TO_TORCH_DTYPE.items()}


def dtype_bytes(dtype: torch.dtype) -> int:
    try:
        return _DTYPE_BYTES[dtype]
    except KeyError:
        raise TypeError(f"unsupported column dtype: {dtype}") from None


@dataclass(frozen=True)
class ColumnSpec:
    name: str
    dtype: torch.dtype
    # elements per row (1 for scalar columns, >1 for fixed-shape vector cols)
    numel: int = 1

    @property
    def row_bytes(self) -> int:
        return self.numel * dtype_bytes(self.dtype)


@dataclass
class Schema:
    """Ordered column specs + the derived packed-row layout."""

    columns: List[ColumnSpec]
    # Derived packed layout (filled in __post_init__):
    packed_order: List[int] = field(default_factory=list)  # idx into columns
    offsets: Dict[str, int] = field(default_factory=dict)  # byte offset in row
    row_stride: int = 0  # padded row byte stride

    def __post_init__(self):
        names = [c.name for c in self.columns]
        if len(set(names)) != len(names):
            raise ValueError(f"duplicate column names in schema: {names}")
        # Descending dtype size => every offset naturally aligned.
        self.packed_order = sorted(
            range(len(self.columns)),
            key=lambda i: (-dtype_bytes(self.columns[i].dtype), i),
        )
        off = 0
        self.offsets = {}
        for i in self.packed_order:
            col = self.columns[i]
            self.offsets[col.name] = off
            off += col.row_bytes
        self.payload_bytes = off
        self.row_stride = (off + 15) // 16 * 16  # 16-B padded for dwordx4

    def __len__(self) -> int:
        return len(self.columns)

    @property
    def names(self) -> List[str]:
        return [c.name for c in self.columns]

    def col(self, name: str) -> ColumnSpec:
        for c in self.columns:
            if c.name == name:
                return c
        raise KeyError(name)

    @staticmethod
    def from_columns(
        columns: Dict[str, torch.Tensor],
    ) -> "Schema":
        specs = []
        for name, t in columns.items():
            numel = 1 if t.dim() == 1 else int(np.prod(t.shape[1:]))
            specs.append(ColumnSpec(name, t.dtype, numel))
        return Schema(specs)

    @staticmethod
    def from_arrow(arrow_schema) -> "Schema":
        """Build from a pyarrow schema (fixed-width columns only)."""
        specs = []
        for fld in arrow_schema:
            np_dt = np.dtype(fld.type.to_pandas_dtype())
            specs.append(ColumnSpec(fld.name, NUMPY_TO_TORCH_DTYPE[np_dt], 1))
        return Schema(specs)


def homogeneous_dtype(schema: Schema) -> Optional[torch.dtype]:
    """If all columns share one dtype, return it (enables the zero-copy
    packed<->columnar fast path); else None."""
    dts = {c.dtype for c in schema.columns}
    return dts.pop() if len(dts) == 1 else None
