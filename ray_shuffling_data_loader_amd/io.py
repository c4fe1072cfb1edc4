"""Parquet ingest: files -> packed row tensors.

Replaces the reference's ``pd.read_parquet`` map-stage load (reference
shuffle.py:151) with multi-threaded pyarrow reads (host C++ Arrow decode,
GIL-released) that land in torch tensors and are packed into the row-major
exchange layout — on GPU via the pack_columns HIP kernel when a device is
given, else via numpy strided views.
"""

import warnings
from concurrent.futures import ThreadPoolExecutor
from typing import Dict, List, Optional, Sequence, Tuple

import numpy as np
import pyarrow.parquet as pq
import torch

from ray_shuffling_data_loader_amd.ops.shuffle_ops import pack_columns
from ray_shuffling_data_loader_amd.utils.schema import (
    ColumnSpec,
    NUMPY_TO_TORCH_DTYPE,
    Schema,
)


def infer_schema(filename: str, drop_columns: Sequence[str] = ()) -> Schema:
    """Schema from a Parquet file's arrow schema (fixed-width columns)."""
    pf = pq.ParquetFile(filename)
    specs = []
    for fld in pf.schema_arrow:
        if fld.name in drop_columns:
            continue
        np_dt = np.dtype(fld.type.to_pandas_dtype())
        specs.append(ColumnSpec(fld.name, NUMPY_TO_TORCH_DTYPE[np_dt], 1))
    return Schema(specs)


def fuse_schema(base: Schema, feature_matrix: Optional[Tuple[str, List[str]]]
                ) -> Schema:
    """Fuse a run of same-dtype scalar columns that are CONSECUTIVE in the
    packed layout into one vector column (e.g. 100 float32 feature columns
    -> one [N,100] 'features' matrix). Offsets must already be contiguous —
    this is a reinterpretation of the packed bytes, not a repack."""
    if feature_matrix is None:
        return base
    name, members = feature_matrix
    dts = {base.col(m).dtype for m in members}
    if len(dts) != 1:
        raise ValueError("feature-matrix columns must share one dtype")
    dt = dts.pop()
    from ray_shuffling_data_loader_amd.utils.schema import dtype_bytes

    offs = sorted(base.offsets[m] for m in members)
    esz = dtype_bytes(dt)
    for a, b in zip(offs, offs[1:]):
        if b - a != esz:
            raise ValueError(
                "feature-matrix columns are not contiguous in packed layout"
            )
    # Declare the fused schema in the BASE's packed-layout order (which is
    # non-increasing in dtype size), collapsing the member run into one
    # vector column in place: the schema's stable (-size, index) sort then
    # reproduces the base layout byte-for-byte. Declaring the fused column
    # first breaks when all columns share a dtype size (e.g. float64
    # features behind an int64 key: the key must stay at offset 0).
    numel = sum(base.col(m).numel for m in members)
    fused_specs = []
    member_set = set(members)
    for i in base.packed_order:
        c = base.columns[i]
        if c.name in member_set:
            if not any(fs.name == name for fs in fused_specs):
                fused_specs.append(ColumnSpec(name, dt, numel))
        else:
            fused_specs.append(c)
    fused = Schema(fused_specs)
    # The fused layout must reinterpret the SAME bytes.
    if fused.offsets[name] != offs[0] or fused.row_stride != base.row_stride:
        raise ValueError(
            "fused schema layout does not match base packed layout "
            f"({fused.offsets[name]} != {offs[0]} or "
            f"{fused.row_stride} != {base.row_stride})"
        )
    return fused


def read_file_columns(
    filename: str,
    schema: Schema,
    pin: bool = False,
) -> Dict[str, torch.Tensor]:
    """Read one Parquet file into contiguous column tensors (only the
    schema's columns). With ``pin=True`` each column is copied ONCE from the
    (read-only) Arrow buffer straight into a pinned-host tensor, ready for
    async H2D — no intermediate writable copy."""
    table = pq.read_table(filename, columns=schema.names)
    out = {}
    for spec in schema.columns:
        arr = table.column(spec.name).to_numpy(zero_copy_only=False)
        if not arr.flags["C_CONTIGUOUS"]:
            arr = np.ascontiguousarray(arr)
        with warnings.catch_warnings():
            # Arrow buffers are read-only; we never write through this
            # tensor (it is copied into pinned memory or consumed by the
            # pack kernel input path), so the writability warning is noise.
            warnings.simplefilter("ignore")
            t = torch.from_numpy(arr)
        if pin:
            p = torch.empty_like(t, pin_memory=True)
            p.copy_(t)
            t = p
        out[spec.name] = t
    return out


def read_files_packed(
    filenames: Sequence[str],
    schema: Schema,
    device: torch.device,
    reader_threads: int = 8,
) -> torch.Tensor:
    """Read + pack many Parquet files into one [N, row_stride] uint8 tensor
    on ``device``. Reads are threaded (Arrow releases the GIL); packing runs
    on the GPU (pack_columns kernel) when device is cuda."""
    if not filenames:
        return torch.empty(0, schema.row_stride, dtype=torch.uint8,
                           device=device)
    use_gpu = device.type == "cuda"

    # Reader threads do Arrow decode + the single copy into pinned-host
    # arenas in parallel; the caller thread only issues async H2D DMAs and
    # pack kernels, which overlap the remaining reads (torch's caching host
    # allocator keeps each pinned block alive until its copy's stream work
    # completes).
    def load(fn):
        return read_file_columns(fn, schema, pin=use_gpu)

    # Preallocate the full source block and pack each file's rows into its
    # slice — no torch.cat, no 2x transient memory at ingest.
    counts = [pq.ParquetFile(fn).metadata.num_rows for fn in filenames]
    total = sum(counts)
    packed = torch.empty(
        total, schema.row_stride, dtype=torch.uint8, device=device
    )
    if device.type == "cpu":
        packed.zero_()  # CPU path packs payload only; keep padding defined
    off = 0
    with ThreadPoolExecutor(max_workers=max(1, reader_threads)) as pool:
        for n_rows, cols_host in zip(counts, pool.map(load, filenames)):
            if use_gpu:
                cols = {
                    name: t.to(device, non_blocking=True)
                    for name, t in cols_host.items()
                }
            else:
                cols = cols_host
            pack_columns(cols, schema, out=packed[off : off + n_rows])
            off += n_rows
    return packed
