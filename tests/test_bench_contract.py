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


def test_bench_world2_gloo(tmp_path):
    """The driver's SCALE tier launches bench.py under torchrun with N>1.
    Exercise that exact path at world 2 on CPU/gloo: per-rank stripe
    generation + barrier, MAX-over-ranks timing reduce, and exactly one
    JSON line on stdout (rank 0 only)."""
    env = dict(os.environ)
    env["RSDL_TUNABLEOP"] = "0"
    proc = subprocess.run(
        [
            sys.executable, "-m", "torch.distributed.run",
            "--standalone", "--local-addr", "127.0.0.1",
            "--nproc-per-node", "2",
            os.path.join(REPO, "bench.py"),
            "--gpus", "2", "--steps", "3", "--warmup", "1",
            "--rows-per-gpu", "80000", "--batch-size", "20000",
            "--num-cols", "8", "--files-per-gpu", "2",
            "--reducers-per-gpu", "2", "--device", "cpu",
            "--dtype", "fp32", "--data-dir", str(tmp_path),
        ],
        capture_output=True,
        text=True,
        timeout=600,
        env=env,
        cwd=REPO,
    )
    assert proc.returncode == 0, proc.stderr[-3000:]
    lines = [ln for ln in proc.stdout.splitlines() if ln.strip()]
    json_lines = [ln for ln in lines if ln.lstrip().startswith("{")]
    assert len(json_lines) == 1, f"exactly one JSON line, got: {lines}"
    d = json.loads(json_lines[0])
    assert d["n_gpus"] == 2
    assert d["config"]["parallelism"] == "dp2"
    assert d["value"] > 0


def test_bench_world4_gloo(tmp_path):
    """Same path at world 4: 4-way all-to-all splits and the MAX-over-4
    timing reduce (closest CPU rehearsal of the driver's 8-GPU tier)."""
    env = dict(os.environ)
    env["RSDL_TUNABLEOP"] = "0"
    proc = subprocess.run(
        [
            sys.executable, "-m", "torch.distributed.run",
            "--standalone", "--local-addr", "127.0.0.1",
            "--nproc-per-node", "4",
            os.path.join(REPO, "bench.py"),
            "--gpus", "4", "--steps", "2", "--warmup", "1",
            "--rows-per-gpu", "40000", "--batch-size", "10000",
            "--num-cols", "8", "--files-per-gpu", "1",
            "--reducers-per-gpu", "1", "--device", "cpu",
            "--dtype", "fp32", "--data-dir", str(tmp_path),
        ],
        capture_output=True,
        text=True,
        timeout=600,
        env=env,
        cwd=REPO,
    )
    assert proc.returncode == 0, proc.stderr[-3000:]
    lines = [ln for ln in proc.stdout.splitlines() if ln.strip()]
    json_lines = [ln for ln in lines if ln.lstrip().startswith("{")]
    assert len(json_lines) == 1, f"exactly one JSON line, got: {lines}"
    d = json.loads(json_lines[0])
    assert d["n_gpus"] == 4
    assert d["config"]["parallelism"] == "dp4"
    assert d["value"] > 0


def test_flat_grad_views_regions():
    """bench.flat_grad_views: every param's .grad is a view of the flat
    buffer, the hook regions are a disjoint cover, and each region holds
    exactly the params fused_step populates at that stage."""
    import sys as _sys

    import torch

    _sys.path.insert(0, REPO)
    import bench
    from ray_shuffling_data_loader_amd.models.mlp import TabularMLP

    model = TabularMLP(100)
    flat, regions = bench.flat_grad_views(model, torch.device("cpu"))
    assert flat.numel() == sum(p.numel() for p in model.parameters())
    # disjoint cover in order w1,w2,w3,bias
    spans = [regions[k] for k in ("w1", "w2", "w3", "bias")]
    assert spans[0][0] == 0 and spans[-1][1] == flat.numel()
    for (a, b), (c, d) in zip(spans, spans[1:]):
        assert b == c
    # writing a param grad shows up in its region only
    named = dict(model.named_parameters())
    stage_of = {
        "net.0.weight": "w1",
        "net.1.weight": "w2",
        "net.2.weight": "w3",
        "net.3.weight": "bias",  # W4 rides with the biases
        "net.0.bias": "bias",
        "net.1.bias": "bias",
        "net.2.bias": "bias",
        "net.3.bias": "bias",
    }
    for name, p in named.items():
        flat.zero_()
        p.grad.fill_(1.0)
        s, e = regions[stage_of[name]]
        assert flat[s:e].abs().sum() == p.numel(), name
        assert flat.abs().sum() == p.numel(), name
