"""Tests for the stats collectors, memory sampler and CSV reports
(reference stats.py parity), plus io schema fusion."""

import csv
import os

import time

import pytest

from ray_shuffling_data_loader_amd.utils.stats import (
    CONSUMER_FIELDS,
    EPOCH_FIELDS,
    TRIAL_FIELDS,
    MemoryStatsCollector,
    TrialStatsCollector,
    get_memory_sample,
    human_readable_big_num,
    human_readable_size,
    process_stats,
)


def test_human_readable():
    assert human_readable_size(1024) == "1.0KiB"
    assert human_readable_size(1536, precision=1) == "1.5KiB"
    assert human_readable_big_num(250000) == "250K"
    assert human_readable_big_num(4 * 10**8) == "400M"


def test_trial_stats_collector_flow():
    c = TrialStatsCollector(
        num_epochs=2, num_maps=3, num_reduces=4, num_consumes=1
    )
    for epoch in range(2):
        c.epoch_start(epoch)
        c.epoch_throttle_done(epoch, 0.01)
        for _ in range(3):
            c.map_start(epoch)
        for _ in range(3):
            c.map_done(epoch, 0.1, 0.05)
        for _ in range(4):
            c.reduce_start(epoch)
            c.reduce_done(epoch, 0.2)
            c.consume_batch(epoch, 1000)
        c.consume_done(epoch)
    c.trial_done(1.5)
    stats = c.get_stats(timeout=5)
    assert stats.duration == 1.5
    assert len(stats.epoch_stats) == 2
    e = stats.epoch_stats[0]
    assert len(e.map_stats.task_durations) == 3
    assert len(e.map_stats.read_durations) == 3
    assert len(e.reduce_stats.task_durations) == 4
    assert e.map_stats.stage_duration >= 0
    assert e.throttle_stats.wait_duration == 0.01
    assert len(e.consume_stats.time_to_consumes) == 4


def test_memory_sampler():
    with MemoryStatsCollector(sample_period_s=0.05) as mem:
        time.sleep(0.2)
    assert len(mem.samples) >= 2
    t, s = mem.samples[0]
    assert s.object_store_bytes_used >= 0


def test_process_stats_csvs(tmp_path):
    c = TrialStatsCollector(1, 2, 2, 1)
    c.epoch_start(0)
    c.map_start(0)
    c.map_start(0)
    c.map_done(0, 0.1, 0.05)
    c.map_done(0, 0.1, 0.05)
    c.reduce_start(0)
    c.reduce_done(0, 0.1)
    c.consume_batch(0, 500)
    c.reduce_start(0)
    c.reduce_done(0, 0.1)
    c.consume_batch(0, 500)
    c.consume_done(0)
    c.trial_done(0.7)
    trial = c.get_stats(timeout=5)
    samples = [(0.0, get_memory_sample())]
    process_stats(
        [(trial, samples)],
        overwrite_stats=True,
        stats_dir=str(tmp_path),
        no_epoch_stats=False,
        no_consumer_stats=False,
        unique_stats=False,
        num_rows=1000,
        num_files=2,
        num_row_groups_per_file=1,
        batch_size=100,
        num_reducers=2,
        num_trainers=1,
        num_epochs=1,
        max_concurrent_epochs=1,
    )
    files = sorted(os.listdir(tmp_path))
    assert len(files) == 3
    for f in files:
        with open(tmp_path / f) as fh:
            rows = list(csv.DictReader(fh))
        assert rows, f
        if f.startswith("trial_stats"):
            assert set(rows[0].keys()) == set(TRIAL_FIELDS)
            assert float(rows[0]["row_throughput"]) == pytest.approx(
                1000 / 0.7
            )
        elif f.startswith("epoch_stats"):
            assert set(rows[0].keys()) == set(EPOCH_FIELDS)
        else:
            assert set(rows[0].keys()) == set(CONSUMER_FIELDS)


def test_fuse_schema(tmp_path):
    from ray_shuffling_data_loader_amd.data_generation import (
        float_data_spec,
        generate_data,
    )
    from ray_shuffling_data_loader_amd.io import fuse_schema, infer_schema

    filenames, _ = generate_data(
        100, 1, 1, 0.0, str(tmp_path), spec=float_data_spec(8),
        include_key=False,
    )
    base = infer_schema(filenames[0])
    fused = fuse_schema(base, ("x", [f"f{i}" for i in range(8)]))
    assert fused.col("x").numel == 8
    assert fused.offsets["x"] == base.offsets["f0"]
    assert fused.row_stride == base.row_stride
    # Non-contiguous member set must be rejected.
    with pytest.raises(ValueError):
        fuse_schema(base, ("x", ["f0", "f2"]))


def test_benchmark_harness_smoke(tmp_path):
    # End-to-end harness parity path (SinkConsumer + run_trials + CSVs).
    import subprocess
    import sys

    r = subprocess.run(
        [
            sys.executable,
            "benchmarks/benchmark.py",
            "--num-rows", "20000",
            "--num-files", "4",
            "--num-reducers", "2",
            "--num-trainers", "1",
            "--num-epochs", "2",
            "--num-trials", "1",
            "--batch-size", "1000",
            "--stats-dir", str(tmp_path),
            "--data-dir", str(tmp_path / "data"),
            "--overwrite-stats",
        ],
        capture_output=True,
        text=True,
        timeout=180,
        cwd=os.path.dirname(
            os.path.dirname(os.path.abspath(__file__))
        ),
    )
    assert r.returncode == 0, r.stderr[-2000:]
    assert "rows/s" in r.stdout
    assert any(
        f.startswith("trial_stats") for f in os.listdir(tmp_path)
    )
