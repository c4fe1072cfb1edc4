"""Ingest tests: row-group-parallel read_files_packed.

The reference's map stage is one pd.read_parquet per file (reference
shuffle.py:151); our ingest splits read tasks at ROW-GROUP granularity and
packs out of order into disjoint slices. These tests pin down that the
packed block is byte-identical to a naive in-order read regardless of task
split, coalescing, or completion order.
"""

import numpy as np
import pytest
import torch

from ray_shuffling_data_loader_amd.data_generation import (
    float_data_spec,
    generate_data,
)
from ray_shuffling_data_loader_amd.io import (
    infer_schema,
    read_file_columns,
    read_files_packed,
)
from ray_shuffling_data_loader_amd.ops.shuffle_ops import pack_columns


def _make_data(tmp_path, num_rows, num_files, num_row_groups, include_key):
    filenames, _ = generate_data(
        num_rows,
        num_files,
        num_row_groups,
        0.0,
        str(tmp_path),
        spec=float_data_spec(7),
        include_key=include_key,
    )
    return list(filenames)


def _reference_pack(filenames, schema):
    """In-order, single-threaded oracle: per-file read + pack."""
    import pyarrow.parquet as pq

    counts = [pq.ParquetFile(fn).metadata.num_rows for fn in filenames]
    total = sum(counts)
    out = torch.zeros(total, schema.row_stride, dtype=torch.uint8)
    off = 0
    for fn, n in zip(filenames, counts):
        cols = read_file_columns(fn, schema)
        pack_columns(cols, schema, out=out[off : off + n])
        off += n
    return out


@pytest.mark.parametrize("num_row_groups", [1, 3, 5])
def test_row_group_parallel_pack_matches_in_order(tmp_path, num_row_groups):
    filenames = _make_data(
        tmp_path, 10_000, 3, num_row_groups, include_key=True
    )
    schema = infer_schema(filenames[0])
    ref = _reference_pack(filenames, schema)
    got = read_files_packed(
        filenames, schema, torch.device("cpu"), reader_threads=4
    )
    assert got.shape == ref.shape
    assert torch.equal(got, ref)


def test_row_group_tasks_not_coalesced_below_min(tmp_path, monkeypatch):
    """With RSDL_MIN_TASK_ROWS=1 every row group is its own task; the packed
    result must still be identical (out-of-order completion safe)."""
    monkeypatch.setenv("RSDL_MIN_TASK_ROWS", "1")
    filenames = _make_data(tmp_path, 9_000, 2, 6, include_key=False)
    schema = infer_schema(filenames[0])
    ref = _reference_pack(filenames, schema)
    got = read_files_packed(
        filenames, schema, torch.device("cpu"), reader_threads=8
    )
    assert torch.equal(got, ref)


def test_read_window_bounds_inflight(tmp_path, monkeypatch):
    """A tiny in-flight window must not change the result."""
    monkeypatch.setenv("RSDL_READ_WINDOW", "1")
    monkeypatch.setenv("RSDL_MIN_TASK_ROWS", "1")
    filenames = _make_data(tmp_path, 6_000, 2, 4, include_key=True)
    schema = infer_schema(filenames[0])
    ref = _reference_pack(filenames, schema)
    got = read_files_packed(
        filenames, schema, torch.device("cpu"), reader_threads=2
    )
    assert torch.equal(got, ref)


def test_empty_filenames():
    from ray_shuffling_data_loader_amd.utils.schema import ColumnSpec, Schema

    schema = Schema([ColumnSpec("a", torch.float32, 1)])
    out = read_files_packed([], schema, torch.device("cpu"))
    assert out.shape == (0, schema.row_stride)


def test_key_column_round_trip(tmp_path):
    """Row identity: the key column survives packing in file order."""
    filenames = _make_data(tmp_path, 5_000, 2, 5, include_key=True)
    schema = infer_schema(filenames[0])
    packed = read_files_packed(
        filenames, schema, torch.device("cpu"), reader_threads=4
    )
    esz = 8
    off = schema.offsets["key"] // esz
    dt = schema.col("key").dtype
    keys = packed.view(dt)[:, off].numpy()
    assert np.array_equal(np.sort(keys), np.arange(5_000))
