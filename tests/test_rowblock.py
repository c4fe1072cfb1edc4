"""Unit tests for RowBlock and Schema construction helpers."""

import numpy as np
import pytest
import torch

from ray_shuffling_data_loader_amd.utils.rowblock import RowBlock
from ray_shuffling_data_loader_amd.utils.schema import ColumnSpec, Schema


def make_block(n=10):
    return RowBlock(
        {
            "a": torch.arange(n, dtype=torch.int64),
            "b": torch.arange(n, dtype=torch.float32) / 10,
            "v": torch.arange(n * 3, dtype=torch.float32).reshape(n, 3),
        }
    )


def test_len_and_columns():
    b = make_block(7)
    assert len(b) == 7
    assert b.names == ["a", "b", "v"]
    assert b.device.type == "cpu"


def test_mismatched_lengths_rejected():
    with pytest.raises(ValueError):
        RowBlock({"a": torch.zeros(3), "b": torch.zeros(4)})
    with pytest.raises(ValueError):
        RowBlock({})


def test_slice_is_view():
    b = make_block(10)
    s = b.slice(2, 5)
    assert len(s) == 3
    assert torch.equal(s["a"], torch.tensor([2, 3, 4]))
    # views share storage
    s["a"][0] = 99
    assert b["a"][2] == 99


def test_slice_clamps():
    b = make_block(5)
    assert len(b.slice(3, 100)) == 2
    assert len(b.slice(0)) == 5
    assert len(b[1:3]) == 2


def test_concat_and_skip_none():
    b1, b2 = make_block(3), make_block(4)
    c = RowBlock.concat([None, b1, b2])
    assert len(c) == 7
    assert torch.equal(c["a"][:3], b1["a"])
    # single block returns as-is (no copy)
    assert RowBlock.concat([None, b1]) is b1
    with pytest.raises(ValueError):
        RowBlock.concat([None])


def test_pandas_roundtrip():
    b = make_block(6)
    df = b.to_pandas()
    assert list(df.columns) == ["a", "b", "v"]
    assert len(df) == 6
    back = RowBlock.from_pandas(df)
    assert torch.equal(back["a"], b["a"])
    assert torch.equal(back["v"], b["v"])


def test_from_numpy():
    b = RowBlock.from_numpy({"x": np.arange(4, dtype=np.float32)})
    assert torch.equal(b["x"], torch.arange(4, dtype=torch.float32))


def test_schema_from_columns():
    cols = {
        "m": torch.zeros(5, 8, dtype=torch.float32),
        "s": torch.zeros(5, dtype=torch.int64),
    }
    sch = Schema.from_columns(cols)
    assert sch.col("m").numel == 8
    assert sch.col("s").dtype == torch.int64
    # int64 packs first (descending size)
    assert sch.offsets["s"] == 0
    assert sch.offsets["m"] == 8


def test_schema_rejects_duplicates():
    with pytest.raises(ValueError):
        Schema([ColumnSpec("x", torch.float32), ColumnSpec("x", torch.int64)])


def test_schema_unsupported_dtype():
    with pytest.raises(TypeError):
        Schema([ColumnSpec("c", torch.complex64)])


def test_fuse_schema_uniform_dtype_sizes():
    """Fusing float64 feature columns behind an int64 key keeps the base
    packed layout (regression: declaring the fused column first moved it
    to offset 0 when every column shares a dtype size)."""
    from ray_shuffling_data_loader_amd.io import fuse_schema

    specs = [ColumnSpec("key", torch.int64)]
    specs += [ColumnSpec(f"f{i}", torch.float64) for i in range(4)]
    specs.append(ColumnSpec("labels", torch.float64))
    base = Schema(specs)
    fused = fuse_schema(base, ("__features__", [f"f{i}" for i in range(4)]))
    assert fused.offsets["key"] == base.offsets["key"] == 0
    assert fused.offsets["__features__"] == base.offsets["f0"] == 8
    assert fused.offsets["labels"] == base.offsets["labels"]
    assert fused.row_stride == base.row_stride
    # fp32 features behind an int64 key (the original working case)
    specs32 = [ColumnSpec("key", torch.int64)]
    specs32 += [ColumnSpec(f"f{i}", torch.float32) for i in range(4)]
    specs32.append(ColumnSpec("labels", torch.float32))
    base32 = Schema(specs32)
    fused32 = fuse_schema(
        base32, ("__features__", [f"f{i}" for i in range(4)])
    )
    assert fused32.offsets["__features__"] == base32.offsets["f0"] == 8
    assert fused32.row_stride == base32.row_stride
