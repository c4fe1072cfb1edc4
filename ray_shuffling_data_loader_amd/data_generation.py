"""Synthetic Parquet data generation.

Parity with the reference generator (reference: ray_shuffling_data_loader/
data_generation.py:13-93): same ``DATA_SPEC`` (17 int64 embedding columns,
2 int64 one-hot columns, 1 float64 label, plus a ``key`` index column), same
file/row-group splitting rules, snappy Parquet output. Ray tasks are replaced
by a thread pool (pyarrow's writer releases the GIL), and a second spec
``float_data_spec`` generates the MI355X flagship benchmark shape
(N float32 feature columns + float32 label; BASELINE.json: 1e8 rows x 100
float cols).
"""

import os
from concurrent.futures import ThreadPoolExecutor
from typing import Dict, Optional, Tuple

import numpy as np
import pyarrow as pa
import pyarrow.parquet as pq

DATA_SPEC = {
    "embeddings_name0": (0, 2385, np.int64),
    "embeddings_name1": (0, 201, np.int64),
    "embeddings_name2": (0, 201, np.int64),
    "embeddings_name3": (0, 6, np.int64),
    "embeddings_name4": (0, 19, np.int64),
    "embeddings_name5": (0, 1441, np.int64),
    "embeddings_name6": (0, 201, np.int64),
    "embeddings_name7": (0, 22, np.int64),
    "embeddings_name8": (0, 156, np.int64),
    "embeddings_name9": (0, 1216, np.int64),
    "embeddings_name10": (0, 9216, np.int64),
    "embeddings_name11": (0, 88999, np.int64),
    "embeddings_name12": (0, 941792, np.int64),
    "embeddings_name13": (0, 9405, np.int64),
    "embeddings_name14": (0, 83332, np.int64),
    "embeddings_name15": (0, 828767, np.int64),
    "embeddings_name16": (0, 945195, np.int64),
    "one_hot0": (0, 3, np.int64),
    "one_hot1": (0, 50, np.int64),
    "labels": (0, 1, np.float64),
}


def float_data_spec(
    num_features: int = 100, dtype=np.float64
) -> Dict[str, Tuple]:
    """The flagship benchmark shape: ``num_features`` float feature columns
    + one float label column (BASELINE.json: '1e8 rows x 100 float cols').
    Default float64 source columns to match the reference's DATA_SPEC
    precision (reference data_generation.py DATA_SPEC: float64 features);
    the fused unpack kernel casts fp64 -> fp32/bf16 on the GPU."""
    spec = {f"f{i}": (0.0, 1.0, dtype) for i in range(num_features)}
    spec["labels"] = (0.0, 1.0, dtype)
    return spec


def generate_row_group(
    group_index: int,
    global_row_index: int,
    num_rows_in_group: int,
    spec: Dict[str, Tuple],
    include_key: bool = True,
    rng: Optional[np.random.Generator] = None,
) -> Dict[str, np.ndarray]:
    """One row group of synthetic columns per the spec
    (reference data_generation.py:80-93)."""
    if rng is None:
        rng = np.random.default_rng(
            abs(hash((group_index, global_row_index))) % (2**32)
        )
    buffer: Dict[str, np.ndarray] = {}
    if include_key:
        buffer["key"] = np.arange(
            global_row_index,
            global_row_index + num_rows_in_group,
            dtype=np.int64,
        )
    for col, (low, high, dtype) in spec.items():
        if np.issubdtype(dtype, np.integer):
            buffer[col] = rng.integers(
                low, high, num_rows_in_group, dtype=dtype
            )
        else:
            buffer[col] = (
                (high - low) * rng.random(num_rows_in_group) + low
            ).astype(dtype)
    return buffer


def generate_file(
    file_index: int,
    global_row_index: int,
    num_rows_in_file: int,
    num_row_groups_per_file: int,
    data_dir: str,
    spec: Dict[str, Tuple] = DATA_SPEC,
    include_key: bool = True,
    compression: str = "snappy",
) -> Tuple[str, int]:
    """Write one snappy Parquet file of ``num_rows_in_file`` rows
    (reference data_generation.py:31-53)."""
    rows_per_group = max(1, num_rows_in_file // num_row_groups_per_file)
    tables = []
    for group_index, group_row_index in enumerate(
        range(0, num_rows_in_file, rows_per_group)
    ):
        n = min(rows_per_group, num_rows_in_file - group_row_index)
        cols = generate_row_group(
            group_index,
            global_row_index + group_row_index,
            n,
            spec,
            include_key=include_key,
        )
        tables.append(pa.table(cols))
    table = pa.concat_tables(tables)
    data_size = table.nbytes
    filename = os.path.join(
        data_dir, f"input_data_{file_index}.parquet.snappy"
    )
    pq.write_table(
        table,
        filename,
        compression=compression,
        row_group_size=rows_per_group,
    )
    return filename, data_size


def generate_data(
    num_rows: int,
    num_files: int,
    num_row_groups_per_file: int,
    max_row_group_skew: float,
    data_dir: str,
    spec: Dict[str, Tuple] = DATA_SPEC,
    include_key: bool = True,
    max_workers: Optional[int] = None,
) -> Tuple[Tuple[str, ...], int]:
    """Generate ``num_rows`` rows split over ``num_files`` Parquet files
    (reference data_generation.py:13-27). Returns (filenames, total_bytes)."""
    assert max_row_group_skew == 0.0, "row-group skew generation unsupported"
    os.makedirs(data_dir, exist_ok=True)
    jobs = []
    rows_per_file = max(1, num_rows // num_files)
    with ThreadPoolExecutor(
        max_workers=max_workers or min(num_files, os.cpu_count() or 4)
    ) as pool:
        for file_index, global_row_index in enumerate(
            range(0, num_rows, rows_per_file)
        ):
            num_rows_in_file = min(rows_per_file, num_rows - global_row_index)
            jobs.append(
                pool.submit(
                    generate_file,
                    file_index,
                    global_row_index,
                    num_rows_in_file,
                    num_row_groups_per_file,
                    data_dir,
                    spec,
                    include_key,
                )
            )
        results = [j.result() for j in jobs]
    filenames, data_sizes = zip(*results)
    return filenames, sum(data_sizes)
