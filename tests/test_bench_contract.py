"""Driver-contract test for bench.py: stdout is exactly ONE JSON line with
the required fields, runnable standalone on CPU (tiny config)."""

import json
import os
import subprocess
import sys

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def test_bench_stdout_is_one_json_line(tmp_path):
    env = dict(os.environ)
    env["RSDL_TUNABLEOP"] = "0"
    proc = subprocess.run(
        [
            sys.executable,
            os.path.join(REPO, "bench.py"),
            "--gpus", "1", "--steps", "3", "--warmup", "1",
            "--rows-per-gpu", "120000", "--batch-size", "30000",
            "--num-cols", "8", "--files-per-gpu", "2",
            "--reducers-per-gpu", "2", "--device", "cpu",
            "--dtype", "fp32", "--data-dir", str(tmp_path),
        ],
        capture_output=True,
        text=True,
        timeout=300,
        env=env,
        cwd=REPO,
    )
    assert proc.returncode == 0, proc.stderr[-2000:]
    lines = [ln for ln in proc.stdout.splitlines() if ln.strip()]
    assert len(lines) == 1, f"stdout must be one JSON line, got: {lines}"
    d = json.loads(lines[0])
    for key in (
        "metric", "value", "unit", "n_gpus", "steps", "warmup",
        "ms_per_step", "higher_is_better", "scaling", "vs_baseline",
        "dtype", "data", "config",
    ):
        assert key in d, key
    assert d["metric"] == "shuffled_rows_per_sec"
    assert d["n_gpus"] == 1 and d["steps"] == 3 and d["warmup"] == 1
    assert d["value"] > 0 and d["ms_per_step"] > 0
    assert d["higher_is_better"] is True and d["scaling"] == "weak"
