"""Parquet ingest: files -> packed row tensors.

Replaces the reference's ``pd.read_parquet`` map-stage load (reference
shuffle.py:151) with multi-threaded pyarrow reads (host C++ Arrow decode,
GIL-released) that land in torch tensors and are packed into the row-major
exchange layout — on GPU via the pack_columns HIP kernel when a device is
given, else via numpy strided views.
"""

import os
import warnings
from concurrent.futures import FIRST_COMPLETED, ThreadPoolExecutor, wait
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


def _read_row_groups_columns(
    filename: str,
    groups: List[int],
    schema: Schema,
    pin: bool,
) -> Dict[str, torch.Tensor]:
    """Decode a run of row groups of one file into column tensors (single
    Arrow copy into ONE pinned arena when ``pin`` — per-column pinned
    allocations at row-group task granularity meant thousands of
    hipHostMalloc/H2D calls per ingest). use_threads=False: task
    granularity already saturates the reader pool; nested Arrow threads
    only oversubscribe."""
    pf = pq.ParquetFile(filename)
    table = pf.read_row_groups(
        groups, columns=schema.names, use_threads=False
    )
    arrs = {}
    metas = []  # (name, byte offset, nbytes)
    total = 0
    for spec in schema.columns:
        arr = table.column(spec.name).to_numpy(zero_copy_only=False)
        if not arr.flags["C_CONTIGUOUS"]:
            arr = np.ascontiguousarray(arr)
        arrs[spec.name] = arr
        total = (total + 15) & ~15  # keep every column slice 16-B aligned
        metas.append((spec.name, total, arr.nbytes))
        total += arr.nbytes
    if not pin:
        out = {}
        for name, _, _ in metas:
            with warnings.catch_warnings():
                warnings.simplefilter("ignore")
                out[name] = torch.from_numpy(arrs[name])
        return out
    arena = torch.empty(total, dtype=torch.uint8, pin_memory=True)
    for name, off, nb in metas:
        with warnings.catch_warnings():
            warnings.simplefilter("ignore")
            arena[off : off + nb].copy_(
                torch.from_numpy(arrs[name].view(np.uint8).ravel())
            )
    return {"__arena__": arena, "__metas__": metas}


def _arena_views(
    arena: torch.Tensor, metas, schema: Schema
) -> Dict[str, torch.Tensor]:
    """Typed column views into a (host or device) ingest arena."""
    out = {}
    for name, off, nb in metas:
        spec = schema.col(name)
        t = arena[off : off + nb].view(spec.dtype)
        out[name] = t if spec.numel == 1 else t.view(-1, spec.numel)
    return out


def read_files_packed(
    filenames: Sequence[str],
    schema: Schema,
    device: torch.device,
    reader_threads: int = 8,
) -> torch.Tensor:
    """Read + pack many Parquet files into one [N, row_stride] uint8 tensor
    on ``device``.

    Tasks are **row-group-granular** (the reference's parallelism unit is
    one map task per file, reference shuffle.py:111-114; file-level tasks
    capped effective reader parallelism at files-per-rank threads — the
    round-1 uncached-ingest bottleneck). Each task decodes one run of row
    groups; destination offsets are precomputed from Parquet metadata so
    tasks complete and pack OUT OF ORDER into disjoint slices of the
    preallocated block (no torch.cat, no 2x transient memory).

    In-flight tasks are bounded to ~2x reader_threads (RSDL_READ_WINDOW):
    decoded pinned column sets can never accumulate beyond the window even
    when the consumer side falls behind the readers.

    GPU path: worker threads do Arrow decode + one copy into pinned
    buffers; the caller thread issues async H2D DMAs + pack kernels as each
    task lands (torch's caching host allocator keeps each pinned block
    alive until its copy's stream work completes). CPU path: workers pack
    directly into their disjoint output slices — the pack itself
    parallelizes.
    """
    if not filenames:
        return torch.empty(0, schema.row_stride, dtype=torch.uint8,
                           device=device)
    use_gpu = device.type == "cuda"

    # Task list: (filename, row-group run, dest row offset, n_rows). Runs
    # are split so ~>= 2 tasks per reader thread exist when possible, but
    # never below one row group (Parquet's decode granularity).
    per_file_groups: List[List[int]] = []
    per_file_sizes: List[List[int]] = []
    total = 0
    for fn in filenames:
        md = pq.ParquetFile(fn).metadata
        sizes = [md.row_group(g).num_rows for g in range(md.num_row_groups)]
        per_file_groups.append(list(range(md.num_row_groups)))
        per_file_sizes.append(sizes)
        total += sum(sizes)

    tasks = []  # (filename, [group_ids], dest_off, n_rows)
    off = 0
    for fn, groups, sizes in zip(filenames, per_file_groups, per_file_sizes):
        for g, n in zip(groups, sizes):
            tasks.append((fn, [g], off, n))
            off += n
    # Coalesce adjacent tiny row groups of the same file so task overhead
    # stays negligible (>= ~250k rows per task unless the file is smaller).
    min_rows = int(os.environ.get("RSDL_MIN_TASK_ROWS", "250000"))
    coalesced = []
    for t in tasks:
        if (
            coalesced
            and coalesced[-1][0] == t[0]
            and coalesced[-1][3] < min_rows
            and coalesced[-1][2] + coalesced[-1][3] == t[2]
        ):
            prev = coalesced[-1]
            coalesced[-1] = (prev[0], prev[1] + t[1], prev[2], prev[3] + t[3])
        else:
            coalesced.append(t)
    tasks = coalesced

    packed = torch.empty(
        total, schema.row_stride, dtype=torch.uint8, device=device
    )
    if device.type == "cpu":
        packed.zero_()  # CPU path packs payload only; keep padding defined

    window = int(
        os.environ.get("RSDL_READ_WINDOW", str(2 * max(1, reader_threads)))
    )

    if not use_gpu:
        # CPU: decode AND pack inside the workers (disjoint output slices).
        def load_pack_cpu(task):
            fn, groups, dst, n_rows = task
            cols = _read_row_groups_columns(fn, groups, schema, pin=False)
            pack_columns(cols, schema, out=packed[dst : dst + n_rows])

        with ThreadPoolExecutor(max_workers=max(1, reader_threads)) as pool:
            list(pool.map(load_pack_cpu, tasks))
        return packed

    def load(task):
        fn, groups, dst, n_rows = task
        return dst, n_rows, _read_row_groups_columns(
            fn, groups, schema, pin=True
        )

    with ThreadPoolExecutor(max_workers=max(1, reader_threads)) as pool:
        pending = set()
        it = iter(tasks)
        for task in it:
            pending.add(pool.submit(load, task))
            if len(pending) >= window:
                break
        while pending:
            done, pending = wait(pending, return_when=FIRST_COMPLETED)
            for fut in done:
                dst, n_rows, host = fut.result()
                # ONE async H2D per task (the pinned arena), then typed
                # device views feed the pack kernel.
                d_arena = host["__arena__"].to(device, non_blocking=True)
                cols = _arena_views(d_arena, host["__metas__"], schema)
                pack_columns(cols, schema, out=packed[dst : dst + n_rows])
                nxt = next(it, None)
                if nxt is not None:
                    pending.add(pool.submit(load, nxt))
    return packed
