#!/usr/bin/env python3
"""Probe Parquet ingest strategies (host decode is the uncached-mode bound)."""
import os
import sys
import tempfile
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

from concurrent.futures import ThreadPoolExecutor

import pyarrow as pa
import pyarrow.parquet as pq
import torch

from ray_shuffling_data_loader_amd.data_generation import (
    float_data_spec,
    generate_data,
)
from ray_shuffling_data_loader_amd.io import infer_schema, read_files_packed


def t(fn, label, nbytes):
    t0 = time.perf_counter()
    r = fn()
    el = time.perf_counter() - t0
    print(f"{label:<44} {el:7.2f}s  {nbytes/el/1e9:6.2f} GB/s")
    return r


def main():
    print("arrow cpu_count:", pa.cpu_count(), " io_threads:", pa.io_thread_count(),
          " os cpus:", os.cpu_count())
    d = tempfile.mkdtemp()
    rows = 10_000_000
    for nf, rg in [(4, 1), (16, 4)]:
        sub = os.path.join(d, f"f{nf}")
        fns, nb = generate_data(rows, nf, rg, 0.0, sub,
                                spec=float_data_spec(100), include_key=False)
        fns = list(fns)
        print(f"--- {nf} files, {rg} row groups/file, {nb/1e9:.1f} GB")
        schema = infer_schema(fns[0])
        dev = torch.device("cuda" if torch.cuda.is_available() else "cpu")

        t(lambda: read_files_packed(fns, schema, dev, 8),
          "read_files_packed (threads=8)", nb)
        t(lambda: read_files_packed(fns, schema, dev, 16),
          "read_files_packed (threads=16)", nb)

        def raw_read(use_threads=True, mmap=False):
            def load(fn):
                return pq.read_table(fn, use_threads=use_threads,
                                     memory_map=mmap)
            with ThreadPoolExecutor(max_workers=len(fns)) as p:
                return list(p.map(load, fns))

        t(lambda: raw_read(), "raw read_table (threads)", nb)
        t(lambda: raw_read(mmap=True), "raw read_table (mmap)", nb)

        def rg_parallel():
            jobs = []
            for fn in fns:
                pf = pq.ParquetFile(fn)
                for g in range(pf.num_row_groups):
                    jobs.append((fn, g))
            def load(job):
                fn, g = job
                return pq.ParquetFile(fn).read_row_group(g)
            with ThreadPoolExecutor(max_workers=min(32, len(jobs))) as p:
                return list(p.map(load, jobs))

        t(rg_parallel, "row-group-parallel reads", nb)

        if torch.cuda.is_available():
            torch.cuda.empty_cache()


if __name__ == "__main__":
    main()
