"""Shuffle stats: data model, collector, memory sampler, CSV reports.

Parity with the reference observability layer (reference:
ray_shuffling_data_loader/stats.py): same dataclasses (StageStats/MapStats/
ReduceStats/ConsumeStats/ThrottleStats/EpochStats/TrialStats), a
TrialStatsCollector with the same callback surface (fired from inside the
shuffle engine instead of Ray tasks; thread-safe via one lock instead of an
actor event loop), the same three CSV reports with identical field schemas
(stats.py:335-381, 484-516, 591-602), and a memory-utilization sampler whose
MI355X analog of the Ray object store is device HBM (torch.cuda.mem_get_info
/ memory_allocated) + host RSS.
"""

import csv
import datetime
import math
import os
import threading
import time
from dataclasses import dataclass
from typing import Dict, List, Optional, Tuple


def _timer() -> float:
    return time.perf_counter()


# ----- data model (reference stats.py:24-64) --------------------------------


@dataclass
class StageStats:
    task_durations: List[float]
    stage_duration: float


@dataclass
class MapStats(StageStats):
    read_durations: List[float]


@dataclass
class ReduceStats(StageStats):
    pass


@dataclass
class ConsumeStats:
    stage_duration: float
    consume_times: Dict[float, int]
    time_to_consumes: List[float]


@dataclass
class ThrottleStats:
    wait_duration: float


@dataclass
class EpochStats:
    duration: float
    map_stats: MapStats
    reduce_stats: ReduceStats
    consume_stats: ConsumeStats
    throttle_stats: ThrottleStats


@dataclass
class TrialStats:
    epoch_stats: List[EpochStats]
    duration: float


# ----- collectors (reference stats.py:72-255) -------------------------------


class _EpochStatsCollector:
    """Per-epoch accumulation; stage durations are first-start/last-done
    (reference stats.py:72-206)."""

    def __init__(self, num_maps: int, num_reduces: int, num_consumes: int):
        self._num_maps = num_maps
        self._num_reduces = num_reduces
        self._num_consumes = num_consumes
        self._duration: Optional[float] = None
        self._epoch_start_time: Optional[float] = None
        self._maps_started = 0
        self._maps_done = 0
        self._map_durations: List[float] = []
        self._read_durations: List[float] = []
        self._reduces_started = 0
        self._reduces_done = 0
        self._reduce_durations: List[float] = []
        self._batches_consumed = 0
        self._consumes_done = 0
        self._consume_times: Dict[float, int] = {}
        self._time_to_consumes: List[float] = []
        self._throttle_duration: Optional[float] = None
        self._map_stage_start: Optional[float] = None
        self._reduce_stage_start: Optional[float] = None
        self._consume_stage_start: Optional[float] = None
        self.map_stage_duration: Optional[float] = None
        self.reduce_stage_duration: Optional[float] = None
        self.consume_stage_duration: Optional[float] = None
        self._epoch_done_ev = threading.Event()

    def epoch_start(self):
        self._epoch_start_time = _timer()
        self._consume_times[self._epoch_start_time] = 0

    def map_start(self):
        if self._maps_started == 0:
            self._map_stage_start = _timer()
        self._maps_started += 1

    def map_done(self, duration, read_duration):
        self._maps_done += 1
        self._map_durations.append(duration)
        self._read_durations.append(read_duration)
        if self._maps_done >= self._num_maps:
            self.map_stage_duration = _timer() - self._map_stage_start

    def reduce_start(self):
        if self._reduces_started == 0:
            self._reduce_stage_start = _timer()
        self._reduces_started += 1

    def reduce_done(self, duration):
        self._reduces_done += 1
        self._reduce_durations.append(duration)
        if self._reduces_done >= self._num_reduces:
            end = _timer()
            self.reduce_stage_duration = end - self._reduce_stage_start
            if self._epoch_start_time is not None:
                self._duration = end - self._epoch_start_time
            self._epoch_done_ev.set()

    def consume_batch(self, num_rows):
        t = _timer()
        if self._batches_consumed == 0:
            self._consume_stage_start = t
        self._consume_times[t] = num_rows
        if self._epoch_start_time is not None:
            self._time_to_consumes.append(t - self._epoch_start_time)
        self._batches_consumed += 1

    def consume_done(self):
        self._consumes_done += 1
        if self._consumes_done >= self._num_consumes:
            if self._consume_stage_start is not None:
                self.consume_stage_duration = (
                    _timer() - self._consume_stage_start
                )

    def throttle_done(self, duration):
        self._throttle_duration = duration

    def get_stats(self, timeout: Optional[float] = None) -> EpochStats:
        self._epoch_done_ev.wait(timeout)
        return EpochStats(
            self._duration or 0.0,
            MapStats(
                self._map_durations,
                self.map_stage_duration or 0.0,
                self._read_durations,
            ),
            ReduceStats(
                self._reduce_durations, self.reduce_stage_duration or 0.0
            ),
            ConsumeStats(
                self.consume_stage_duration or 0.0,
                self._consume_times,
                self._time_to_consumes,
            ),
            ThrottleStats(self._throttle_duration or 0.0),
        )


class TrialStatsCollector:
    """Thread-safe trial-level collector (reference stats.py:209-255; the
    actor event loop is replaced by one lock — engine callbacks are
    control-plane-rate)."""

    def __init__(
        self,
        num_epochs: int,
        num_maps: int,
        num_reduces: int,
        num_consumes: int,
    ):
        self._collectors = [
            _EpochStatsCollector(num_maps, num_reduces, num_consumes)
            for _ in range(num_epochs)
        ]
        self._duration: Optional[float] = None
        self._trial_done_ev = threading.Event()
        self._lock = threading.Lock()

    def trial_done(self, duration):
        with self._lock:
            self._duration = duration
        self._trial_done_ev.set()

    def epoch_throttle_done(self, epoch, duration):
        with self._lock:
            self._collectors[epoch].throttle_done(duration)

    def epoch_start(self, epoch):
        with self._lock:
            self._collectors[epoch].epoch_start()

    def epoch_done(self, epoch, duration):
        # Extra hook vs the reference (engine knows its epoch wall time).
        pass

    def map_start(self, epoch):
        with self._lock:
            self._collectors[epoch].map_start()

    def map_done(self, epoch, duration, read_duration):
        with self._lock:
            self._collectors[epoch].map_done(duration, read_duration)

    def reduce_start(self, epoch):
        with self._lock:
            self._collectors[epoch].reduce_start()

    def reduce_done(self, epoch, duration):
        with self._lock:
            self._collectors[epoch].reduce_done(duration)

    def consume_batch(self, epoch, num_rows):
        with self._lock:
            self._collectors[epoch].consume_batch(num_rows)

    def consume_done(self, epoch):
        with self._lock:
            self._collectors[epoch].consume_done()

    def get_stats(self, timeout: Optional[float] = None) -> TrialStats:
        self._trial_done_ev.wait(timeout)
        epoch_stats = [c.get_stats(timeout) for c in self._collectors]
        return TrialStats(epoch_stats, self._duration or 0.0)


# ----- memory utilization sampler (reference stats.py:258-279, 649-699) -----


@dataclass
class MemorySample:
    hbm_bytes_used: int = 0
    hbm_bytes_total: int = 0
    host_rss_bytes: int = 0

    # Name parity with reference store-stats samples.
    @property
    def object_store_bytes_used(self) -> int:
        return self.hbm_bytes_used or self.host_rss_bytes


def get_memory_sample() -> MemorySample:
    s = MemorySample()
    try:
        import torch

        if torch.cuda.is_available():
            free, total = torch.cuda.mem_get_info()
            s.hbm_bytes_used = total - free
            s.hbm_bytes_total = total
    except Exception:
        pass
    try:
        import psutil

        s.host_rss_bytes = psutil.Process().memory_info().rss
    except Exception:
        pass
    return s


class MemoryStatsCollector:
    """Context manager sampling HBM/host memory on a thread (the reference's
    ObjectStoreStatsCollector polled the raylet gRPC memory service every
    5 s; reference stats.py:258-279, 686-699)."""

    def __init__(self, sample_period_s: float = 5.0, do_print: bool = False):
        self.samples: List[Tuple[float, MemorySample]] = []
        self._period = sample_period_s
        self._print = do_print
        self._done = threading.Event()
        self._thread: Optional[threading.Thread] = None

    def _loop(self):
        while True:
            t = _timer()
            sample = get_memory_sample()
            self.samples.append((t, sample))
            if self._print:
                print(
                    f"[mem] hbm={human_readable_size(sample.hbm_bytes_used)}"
                    f" rss={human_readable_size(sample.host_rss_bytes)}"
                )
            if self._done.wait(timeout=self._period):
                return

    def __enter__(self):
        self._done.clear()
        self._thread = threading.Thread(
            target=self._loop, name="rsdl-mem-sampler", daemon=True
        )
        self._thread.start()
        return self

    def __exit__(self, *exc):
        self._done.set()
        if self._thread is not None:
            self._thread.join(timeout=10)
        return False


# Backwards-compatible alias mirroring the reference class name.
ObjectStoreStatsCollector = MemoryStatsCollector


# ----- reports (reference stats.py:287-626) ---------------------------------

UNITS = ["", "K", "M", "B", "T", "Q"]


def human_readable_big_num(num):
    idx = int(math.log10(num) // 3)
    unit = UNITS[idx]
    new_num = num / 10 ** (3 * idx)
    if new_num % 1 == 0:
        return f"{int(new_num)}{unit}"
    return f"{new_num:.1f}{unit}"


def human_readable_size(num, precision=1, suffix="B"):
    for unit in ["", "Ki", "Mi", "Gi", "Ti", "Pi", "Ei", "Zi"]:
        if abs(num) < 1024.0 or unit == "Zi":
            break
        num /= 1024.0
    return f"{num:.{precision}f}{unit}{suffix}"


def _agg(prefix: str, values: List[float]) -> Dict[str, float]:
    import numpy as np

    if not values:
        values = [0.0]
    return {
        f"avg_{prefix}": float(np.mean(values)),
        f"std_{prefix}": float(np.std(values)),
        f"max_{prefix}": float(np.max(values)),
        f"min_{prefix}": float(np.min(values)),
    }


TRIAL_FIELDS = [
    "num_files", "num_row_groups_per_file", "num_reducers", "num_trainers",
    "num_epochs", "max_concurrent_epochs", "trial", "duration",
    "row_throughput", "batch_throughput", "batch_throughput_per_trainer",
    "avg_object_store_utilization", "max_object_store_utilization",
    "avg_epoch_duration", "std_epoch_duration", "max_epoch_duration",
    "min_epoch_duration", "avg_map_stage_duration", "std_map_stage_duration",
    "max_map_stage_duration", "min_map_stage_duration",
    "avg_reduce_stage_duration", "std_reduce_stage_duration",
    "max_reduce_stage_duration", "min_reduce_stage_duration",
    "avg_consume_stage_duration", "std_consume_stage_duration",
    "max_consume_stage_duration", "min_consume_stage_duration",
    "avg_map_task_duration", "std_map_task_duration",
    "max_map_task_duration", "min_map_task_duration", "avg_read_duration",
    "std_read_duration", "max_read_duration", "min_read_duration",
    "avg_reduce_task_duration", "std_reduce_task_duration",
    "max_reduce_task_duration", "min_reduce_task_duration",
    "avg_time_to_consume", "std_time_to_consume", "max_time_to_consume",
    "min_time_to_consume",
]

EPOCH_FIELDS = [
    "num_files", "num_row_groups_per_file", "num_reducers", "num_trainers",
    "num_epochs", "max_concurrent_epochs", "trial", "epoch", "duration",
    "row_throughput", "batch_throughput", "batch_throughput_per_trainer",
    "map_stage_duration", "reduce_stage_duration", "consume_stage_duration",
    "avg_map_task_duration", "std_map_task_duration",
    "max_map_task_duration", "min_map_task_duration", "avg_read_duration",
    "std_read_duration", "max_read_duration", "min_read_duration",
    "avg_reduce_task_duration", "std_reduce_task_duration",
    "max_reduce_task_duration", "min_reduce_task_duration",
    "avg_time_to_consume", "std_time_to_consume", "max_time_to_consume",
    "min_time_to_consume",
]

CONSUMER_FIELDS = [
    "num_files", "num_row_groups_per_file", "num_reducers", "num_trainers",
    "num_epochs", "max_concurrent_epochs", "trial", "epoch", "timestamp",
    "num_rows_in_reducer_batch",
]


def process_stats(
    all_stats,
    overwrite_stats,
    stats_dir,
    no_epoch_stats,
    no_consumer_stats,
    unique_stats,
    num_rows,
    num_files,
    num_row_groups_per_file,
    batch_size,
    num_reducers,
    num_trainers,
    num_epochs,
    max_concurrent_epochs,
):
    """Aggregate trials, print mean throughput, write the three CSV reports
    (reference stats.py:287-626). ``all_stats`` is a list of
    (TrialStats, [(ts, MemorySample), ...]) tuples."""
    import numpy as np

    stats_list, store_stats_list = zip(*all_stats)
    times = [s.duration for s in stats_list]
    mean = np.mean(times)
    std = np.std(times)
    store_bytes_used = [
        getattr(sample, "object_store_bytes_used", 0)
        for trial_samples in store_stats_list
        for _, sample in trial_samples
    ] or [0]
    num_samples = sum(len(ts) for ts in store_stats_list)
    throughput_std = np.std([num_epochs * num_rows / t for t in times])
    batch_throughput_std = np.std(
        [(num_epochs * num_rows / batch_size) / t for t in times]
    )
    print(f"\nMean over {len(times)} trials: {mean:.3f}s +- {std}")
    print(
        f"Mean throughput over {len(times)} trials: "
        f"{num_epochs * num_rows / mean:.2f} rows/s +- {throughput_std:.2f}"
    )
    print(
        f"Mean batch throughput over {len(times)} trials: "
        f"{(num_epochs * num_rows / batch_size) / mean:.2f} batches/s +- "
        f"{batch_throughput_std:.2f}"
    )
    print(
        f"Max device/host memory utilization over {num_samples} samples: "
        f"{human_readable_size(np.max(store_bytes_used))}\n"
    )

    os.makedirs(stats_dir, exist_ok=True)
    write_mode = "w+" if overwrite_stats else "a+"
    hr_num_rows = human_readable_big_num(num_rows)
    hr_batch_size = human_readable_big_num(batch_size)
    now = datetime.datetime.utcnow().isoformat()

    def open_csv(kind, fields):
        filename = f"{kind}_{hr_num_rows}_rows_{hr_batch_size}_batch_size"
        filename += f"_{now}.csv" if unique_stats else ".csv"
        if "://" in stats_dir:
            # Remote stats dir (e.g. s3://bucket/prefix) via fsspec, as the
            # reference does (reference stats.py: fsspec.open on the
            # hr_stats_dir). Header handling: remote appends are not
            # supported, so always write the header.
            import fsspec

            filename = stats_dir.rstrip("/") + "/" + filename
            # Object stores can't append; always (over)write remote CSVs.
            f = fsspec.open(filename, "w").open()
            writer = csv.DictWriter(f, fieldnames=fields)
            writer.writeheader()
            print(f"Writing out {kind} to {filename}.")
            return f, writer
        filename = os.path.join(stats_dir, filename)
        write_header = (
            overwrite_stats
            or not os.path.exists(filename)
            or os.path.getsize(filename) == 0
        )
        f = open(filename, write_mode)
        writer = csv.DictWriter(f, fieldnames=fields)
        if write_header:
            writer.writeheader()
        print(f"Writing out {kind} to {filename}.")
        return f, writer

    base = {
        "num_files": num_files,
        "num_row_groups_per_file": num_row_groups_per_file,
        "num_reducers": num_reducers,
        "num_trainers": num_trainers,
        "num_epochs": num_epochs,
        "max_concurrent_epochs": max_concurrent_epochs,
    }

    f, writer = open_csv("trial_stats", TRIAL_FIELDS)
    with f:
        for trial, (stats, store_stats) in enumerate(all_stats):
            row = dict(base)
            row["trial"] = trial
            row["duration"] = stats.duration
            row_tp = num_epochs * num_rows / stats.duration
            row["row_throughput"] = row_tp
            row["batch_throughput"] = row_tp / batch_size
            row["batch_throughput_per_trainer"] = (
                row_tp / batch_size / num_trainers
            )
            used = [
                getattr(s, "object_store_bytes_used", 0)
                for _, s in store_stats
            ] or [0]
            row["avg_object_store_utilization"] = float(np.mean(used))
            row["max_object_store_utilization"] = float(np.max(used))
            epochs = stats.epoch_stats
            row.update(_agg("epoch_duration", [e.duration for e in epochs]))
            row.update(
                _agg(
                    "map_stage_duration",
                    [e.map_stats.stage_duration for e in epochs],
                )
            )
            row.update(
                _agg(
                    "reduce_stage_duration",
                    [e.reduce_stats.stage_duration for e in epochs],
                )
            )
            row.update(
                _agg(
                    "consume_stage_duration",
                    [e.consume_stats.stage_duration for e in epochs],
                )
            )
            row.update(
                _agg(
                    "map_task_duration",
                    [d for e in epochs for d in e.map_stats.task_durations],
                )
            )
            row.update(
                _agg(
                    "read_duration",
                    [d for e in epochs for d in e.map_stats.read_durations],
                )
            )
            row.update(
                _agg(
                    "reduce_task_duration",
                    [
                        d
                        for e in epochs
                        for d in e.reduce_stats.task_durations
                    ],
                )
            )
            row.update(
                _agg(
                    "time_to_consume",
                    [
                        d
                        for e in epochs
                        for d in e.consume_stats.time_to_consumes
                    ],
                )
            )
            writer.writerow(row)

    if not no_epoch_stats:
        f, writer = open_csv("epoch_stats", EPOCH_FIELDS)
        with f:
            for trial, (stats, _) in enumerate(all_stats):
                for epoch, e in enumerate(stats.epoch_stats):
                    row = dict(base)
                    row["trial"] = trial
                    row["epoch"] = epoch
                    row["duration"] = e.duration
                    tp = num_rows / e.duration if e.duration else 0.0
                    row["row_throughput"] = tp
                    row["batch_throughput"] = tp / batch_size
                    row["batch_throughput_per_trainer"] = (
                        tp / batch_size / num_trainers
                    )
                    row["map_stage_duration"] = e.map_stats.stage_duration
                    row["reduce_stage_duration"] = (
                        e.reduce_stats.stage_duration
                    )
                    row["consume_stage_duration"] = (
                        e.consume_stats.stage_duration
                    )
                    row.update(
                        _agg("map_task_duration", e.map_stats.task_durations)
                    )
                    row.update(
                        _agg("read_duration", e.map_stats.read_durations)
                    )
                    row.update(
                        _agg(
                            "reduce_task_duration",
                            e.reduce_stats.task_durations,
                        )
                    )
                    row.update(
                        _agg(
                            "time_to_consume",
                            e.consume_stats.time_to_consumes,
                        )
                    )
                    writer.writerow(row)

    if not no_consumer_stats:
        f, writer = open_csv("consumer_stats", CONSUMER_FIELDS)
        with f:
            for trial, (stats, _) in enumerate(all_stats):
                for epoch, e in enumerate(stats.epoch_stats):
                    for ts, n in e.consume_stats.consume_times.items():
                        row = dict(base)
                        row["trial"] = trial
                        row["epoch"] = epoch
                        row["timestamp"] = ts
                        row["num_rows_in_reducer_batch"] = n
                        writer.writerow(row)
