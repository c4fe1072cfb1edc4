#!/usr/bin/env python3
"""Shuffle benchmark harness.

Parity with the reference harness (reference: benchmarks/benchmark.py):
same CLI flags, a no-op sink consumer per trainer with its own epoch window
(reference Consumer actor, benchmark.py:29-62), ``run_trials`` over N trials
or a timeout (benchmark.py:111-184), TrialStatsCollector + memory sampler,
and the three CSV reports with the reference's field schemas
(utils/stats.process_stats). The Ray placement group / cluster plumbing is
replaced by the engine's process-local (1 node, N GPUs) fabric.

Measures the LOADER itself (no model): rows/s with the no-op sink is the
reference's own headline methodology (its benchmark Consumer just logs and
drops the batches, benchmark.py:47-50).
"""

import argparse
import os
import sys
import tempfile
import threading
import time

sys.path.insert(
    0, os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
)

from ray_shuffling_data_loader_amd.batch_queue import BatchQueue
from ray_shuffling_data_loader_amd.data_generation import (
    DATA_SPEC,
    float_data_spec,
    generate_data,
)
from ray_shuffling_data_loader_amd.shuffle import BatchConsumer, shuffle
from ray_shuffling_data_loader_amd.utils.stats import (
    MemoryStatsCollector,
    TrialStatsCollector,
    human_readable_size,
    process_stats,
)


class SinkConsumer(BatchConsumer):
    """Per-trainer no-op sink with the consumer-side epoch window
    (reference benchmark.py:29-62: Consumer actor + its max_concurrent
    window). Batches are counted for stats and dropped."""

    def __init__(
        self,
        num_epochs: int,
        num_trainers: int,
        max_concurrent_epochs: int,
        stats_collector=None,
        consume_latency_s: float = 0.0,
    ):
        self._queue = BatchQueue(
            num_epochs, num_trainers, max_concurrent_epochs
        )
        self._stats = stats_collector
        self._latency = consume_latency_s
        self._threads = [
            threading.Thread(
                target=self._drain, args=(t, num_epochs), daemon=True
            )
            for t in range(num_trainers)
        ]
        for t in self._threads:
            t.start()

    def _drain(self, trainer: int, num_epochs: int):
        for epoch in range(num_epochs):
            while True:
                items = self._queue.get_batch(trainer, epoch)
                done = bool(items) and items[-1] is None
                if done:
                    items.pop()
                for block in items:
                    if self._latency:
                        time.sleep(self._latency)
                    if self._stats:
                        self._stats.consume_batch(epoch, len(block))
                n_ack = len(items) + (1 if done else 0)
                if n_ack:
                    self._queue.task_done(trainer, epoch, n_ack)
                if done:
                    break
            if self._stats:
                self._stats.consume_done(epoch)

    # BatchConsumer interface (producer side) ------------------------------
    def consume(self, rank, epoch, batches):
        self._queue.put_batch(rank, epoch, batches)

    def producer_done(self, rank, epoch):
        self._queue.producer_done(rank, epoch)

    def wait_until_ready(self, epoch):
        self._queue.new_epoch(epoch)

    def wait_until_all_epochs_done(self):
        self._queue.wait_until_all_epochs_done()
        for t in self._threads:
            t.join(timeout=60)


def run_trial(args, filenames, trial_idx):
    stats = TrialStatsCollector(
        args.num_epochs,
        num_maps=args.num_files,
        num_reduces=args.num_reducers,
        num_consumes=args.num_trainers,
    )
    sink = SinkConsumer(
        args.num_epochs,
        args.num_trainers,
        args.max_concurrent_epochs,
        stats_collector=stats,
        consume_latency_s=args.consume_latency,
    )
    mem = MemoryStatsCollector(
        sample_period_s=args.utilization_sample_period
    )
    with mem:
        duration = shuffle(
            filenames,
            sink,
            args.num_epochs,
            args.num_reducers,
            args.num_trainers,
            stats_collector=stats,
            source_cache=args.source_cache,
            reader_threads=args.reader_threads,
            seed=args.seed + trial_idx if args.seed is not None else None,
        )
    print(f"Trial {trial_idx} done in {duration:.3f}s")
    trial_stats = stats.get_stats(timeout=60)
    return trial_stats, list(mem.samples)


def run_trials(args, filenames):
    """N trials or run until timeout (reference benchmark.py:111-184)."""
    all_stats = []
    if args.trials_timeout is not None:
        start = time.perf_counter()
        trial = 0
        while time.perf_counter() - start < args.trials_timeout:
            all_stats.append(run_trial(args, filenames, trial))
            trial += 1
    else:
        for trial in range(args.num_trials):
            all_stats.append(run_trial(args, filenames, trial))
    return all_stats


def parse_args():
    p = argparse.ArgumentParser(description="Shuffling data loader benchmark")
    p.add_argument("--num-rows", type=int, default=4 * 10**6)
    p.add_argument("--num-files", type=int, default=100)
    p.add_argument("--num-row-groups-per-file", type=int, default=5)
    p.add_argument("--num-reducers", type=int, default=5)
    p.add_argument("--num-trainers", type=int, default=5)
    p.add_argument("--num-epochs", type=int, default=10)
    p.add_argument("--max-concurrent-epochs", type=int, default=2)
    p.add_argument("--batch-size", type=int, default=100)
    p.add_argument("--num-trials", type=int, default=3)
    p.add_argument("--trials-timeout", type=int, default=None)
    p.add_argument("--data-dir", type=str, default=None)
    p.add_argument("--stats-dir", type=str, default="./results")
    p.add_argument("--overwrite-stats", action="store_true")
    p.add_argument("--unique-stats", action="store_true")
    p.add_argument("--no-epoch-stats", action="store_true")
    p.add_argument("--no-consumer-stats", action="store_true")
    p.add_argument("--use-old-data", action="store_true")
    p.add_argument("--consume-latency", type=float, default=0.0,
                   help="simulated per-batch consume latency (s)")
    p.add_argument("--utilization-sample-period", type=float, default=5.0)
    p.add_argument("--data-spec", choices=["tabular", "float100"],
                   default="tabular",
                   help="tabular = reference DATA_SPEC; float100 = the "
                   "MI355X flagship 100-float32-column shape")
    p.add_argument("--num-float-cols", type=int, default=100)
    p.add_argument("--reader-threads", type=int, default=8,
                   help="ingest decode threads per engine")
    p.add_argument("--source-cache", type=str, default="none",
                   choices=["auto", "device", "host", "none"],
                   help="'none' re-reads Parquet every epoch like the "
                   "reference; 'auto' keeps the packed dataset HBM-resident")
    p.add_argument("--seed", type=int, default=None)
    return p.parse_args()


def main():
    args = parse_args()
    data_dir = args.data_dir or os.path.join(
        tempfile.gettempdir(), "rsdl_benchmark_data"
    )
    spec = (
        DATA_SPEC
        if args.data_spec == "tabular"
        else float_data_spec(args.num_float_cols)
    )
    marker = os.path.join(
        data_dir,
        f"{args.data_spec}_{args.num_rows}_{args.num_files}",
    )
    if args.use_old_data and os.path.isdir(marker) and os.listdir(marker):
        filenames = sorted(
            os.path.join(marker, f) for f in os.listdir(marker)
        )
        print(f"Reusing {len(filenames)} files in {marker}")
    else:
        print(
            f"Generating {args.num_rows} rows over {args.num_files} files "
            f"({args.num_row_groups_per_file} row groups/file)"
        )
        t0 = time.perf_counter()
        filenames, num_bytes = generate_data(
            args.num_rows,
            args.num_files,
            args.num_row_groups_per_file,
            0.0,
            marker,
            spec=spec,
            include_key=args.data_spec == "tabular",
        )
        filenames = list(filenames)
        print(
            f"Generated {human_readable_size(num_bytes)} in "
            f"{time.perf_counter() - t0:.1f}s"
        )

    all_stats = run_trials(args, filenames)

    process_stats(
        all_stats,
        args.overwrite_stats,
        args.stats_dir,
        args.no_epoch_stats,
        args.no_consumer_stats,
        args.unique_stats,
        args.num_rows,
        args.num_files,
        args.num_row_groups_per_file,
        args.batch_size,
        args.num_reducers,
        args.num_trainers,
        args.num_epochs,
        args.max_concurrent_epochs,
    )


if __name__ == "__main__":
    main()
