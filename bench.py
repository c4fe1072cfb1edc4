#!/usr/bin/env python3
"""Flagship benchmark: shuffled rows/sec through the MI355X shuffling data
loader feeding a real DDP training step.

BASELINE.json metric: "shuffled rows/sec (whole node) + p50 batch-wait,
1e8 x 100 float cols, 8 trainers". Weak scaling: 1.25e7 rows x 100 float64
cols per GPU (N=8 => the named 1e8-row config), batch_size 250k, synthetic
Parquet (no network; generated locally on first run), random-init TabularMLP
with genuine fwd+bwd+optimizer in every timed step (the reference example
mocks its train step with sleep; we do real work on top of the loader).

One step = consume one 250k-row shuffled batch + train on it. The shuffle
pipeline (per-epoch map -> RCCL all-to-all over xGMI -> fused HIP
permute/pack) runs concurrently under max_concurrent_epochs=2.

Launch (driver contract):
  python bench.py --gpus N --steps K --warmup W          # N=1
  python -m torch.distributed.run --nnodes=1 --nproc-per-node N \
      --master-addr 127.0.0.1 bench.py --gpus N ...      # N>1, one rank/GPU
"""

import argparse
import json
import os
import shutil
import statistics
import sys
import tempfile
import time

# hipBLASLt algorithm selection: load the pre-tuned gfx950 GEMM table
# (profiles/tunableop_gfx950.csv — picks split-K kernels for the huge-K
# wgrad shapes; ~20% step time) and keep tuning enabled for any shape not
# in it (tuning runs during warmup). Must be set before torch loads blaslt.
_repo = os.path.dirname(os.path.abspath(__file__))
_tuned_src = os.path.join(_repo, "profiles", "tunableop_gfx950.csv")
if os.environ.get("RSDL_TUNABLEOP", "1") == "1":
    _tuned_dst = os.path.join(
        tempfile.gettempdir(), f"rsdl_tunableop_{os.getpid()}.csv"
    )
    if os.path.exists(_tuned_src):
        # TunableOp resolves FILENAME with the device ordinal inserted
        # before the extension (e.g. foo.csv -> foo0.csv) on both read and
        # write; provide every per-ordinal name so each rank's process
        # actually LOADS the table instead of silently re-tuning in warmup.
        shutil.copyfile(_tuned_src, _tuned_dst)
        _stem, _ext = os.path.splitext(_tuned_dst)
        for _ord in range(8):
            shutil.copyfile(_tuned_src, f"{_stem}{_ord}{_ext}")
    os.environ.setdefault("PYTORCH_TUNABLEOP_ENABLED", "1")
    os.environ.setdefault("PYTORCH_TUNABLEOP_TUNING", "1")
    os.environ.setdefault("PYTORCH_TUNABLEOP_FILENAME", _tuned_dst)
    os.environ.setdefault("PYTORCH_TUNABLEOP_VERBOSE", "0")

# First-contact diagnostics for multi-GPU runs: RCCL warnings/errors go to
# stderr (NCCL_DEBUG must be set before the communicator is created), and a
# watchdog dumps all ranks' Python stacks if a step wedges (RCCL hangs
# surface as silent stalls, not exceptions). RSDL_NCCL_DEBUG=INFO for full
# per-collective logs; RSDL_WATCHDOG_S=0 disables the stack dump.
os.environ.setdefault(
    "NCCL_DEBUG", os.environ.get("RSDL_NCCL_DEBUG", "WARN")
)

import torch

from ray_shuffling_data_loader_amd.data_generation import float_data_spec
from ray_shuffling_data_loader_amd.parallel import fabric
from ray_shuffling_data_loader_amd.models.mlp import TabularMLP
from ray_shuffling_data_loader_amd.torch_dataset import TorchShufflingDataset


def flat_grad_views(model, device):
    """Pre-create every param's .grad as a view of ONE flat buffer, laid
    out [W1, W2, W3, W4, b1..b4] so each of fused_step's grad_hook stages
    ("w1"/"w2"/"w3" after the corresponding wgrad kernel, "bias" = W4 +
    every bias right after the backward chain) maps to ONE contiguous
    slice. Returns (flat, regions). Order is irrelevant to the default
    single-collective path."""
    weights = [p for n, p in model.named_parameters()
               if n.endswith("weight")]
    biases = [p for n, p in model.named_parameters()
              if n.endswith("bias")]
    ordered = weights + biases
    flat = torch.zeros(sum(p.numel() for p in ordered), device=device)
    off = 0
    offs = []
    for p in ordered:
        p.grad = flat[off : off + p.numel()].view_as(p)
        offs.append(off)
        off += p.numel()
    regions = {
        "w1": (offs[0], offs[1]),
        "w2": (offs[1], offs[2]),
        "w3": (offs[2], offs[3]),
        "bias": (offs[3], off),  # W4 + b1..b4
    }
    return flat, regions


def parse_args():
    p = argparse.ArgumentParser()
    p.add_argument("--gpus", type=int, default=1)
    p.add_argument("--steps", type=int, default=80)
    p.add_argument("--warmup", type=int, default=16)
    p.add_argument("--batch-size", type=int, default=250_000)
    p.add_argument("--rows-per-gpu", type=int, default=12_500_000)
    p.add_argument("--num-cols", type=int, default=100)
    p.add_argument("--files-per-gpu", type=int, default=4)
    p.add_argument("--reducers-per-gpu", type=int, default=4)
    p.add_argument("--max-concurrent-epochs", type=int, default=2)
    p.add_argument("--data-dir", type=str, default=None)
    p.add_argument(
        "--device", type=str, default=None, help="override (cpu for debug)"
    )
    p.add_argument(
        "--dtype",
        type=str,
        default="bf16",
        choices=["bf16", "fp32"],
        help="train-step compute dtype; bf16 uses MFMA via autocast and the "
        "loader's fused fp32->bf16 cast kernel for features",
    )
    p.add_argument(
        "--loader-output",
        type=str,
        default="auto",
        choices=["auto", "views", "columns"],
        help="views: zero-copy strided feature matrix; columns: fused "
        "unpack kernel -> contiguous feature matrix (better GEMM input)",
    )
    p.add_argument(
        "--source-cache",
        type=str,
        default="auto",
        choices=["auto", "device", "host", "none"],
        help="auto: dataset HBM-resident after first read",
    )
    return p.parse_args()


def main():
    args = parse_args()
    world, rank = fabric.init_from_env()
    n_gpus = max(args.gpus, world)
    if args.device:
        device = torch.device(args.device)
    elif torch.cuda.is_available():
        device = torch.device("cuda", rank % torch.cuda.device_count())
        torch.cuda.set_device(device)
    else:
        device = torch.device("cpu")

    # ----- synthetic data shard for this rank ------------------------------
    data_dir = args.data_dir or os.path.join(
        tempfile.gettempdir(), "rsdl_bench_data"
    )
    spec = float_data_spec(args.num_cols)
    total_files = args.files_per_gpu * world
    rows_per_file = args.rows_per_gpu // args.files_per_gpu
    shard_dir = os.path.join(
        data_dir, f"w{world}_{args.rows_per_gpu}x{args.num_cols}f64"
    )
    os.makedirs(shard_dir, exist_ok=True)
    filenames = [
        os.path.join(shard_dir, f"input_data_{i}.parquet.snappy")
        for i in range(total_files)
    ]
    # Each rank generates its stripe (engine shards files[rank::world]).
    my_indices = list(range(rank, total_files, world))
    missing = [i for i in my_indices if not os.path.exists(filenames[i])]
    if missing:
        t0 = time.perf_counter()
        from concurrent.futures import ThreadPoolExecutor

        from ray_shuffling_data_loader_amd.data_generation import (
            generate_file,
        )

        with ThreadPoolExecutor(max_workers=min(8, len(missing))) as pool:
            list(
                pool.map(
                    lambda i: generate_file(
                        i,
                        i * rows_per_file,
                        rows_per_file,
                        # ~390k-row row groups: the ingest parallelism unit
                        # (row-group-granular read tasks) — one giant row
                        # group per file would serialize uncached decode.
                        max(1, rows_per_file // 390_625),
                        shard_dir,
                        spec=spec,
                        include_key=False,
                    ),
                    missing,
                )
            )
        # stderr: stdout carries exactly ONE JSON line (driver contract).
        print(
            f"[bench] rank {rank} generated {len(missing)} files "
            f"({len(missing) * rows_per_file} rows) in "
            f"{time.perf_counter() - t0:.1f}s",
            flush=True,
            file=sys.stderr,
        )
    if world > 1:
        torch.distributed.barrier()

    # ----- loader + model --------------------------------------------------
    # The timed window must contain >= 1 epoch rollover so the measured
    # number includes the full per-epoch reshuffle the metric claims (the
    # pipelined epoch e+1 shuffle must actually be ready when epoch e runs
    # out). After warmup we burn extra UNTIMED steps until the epoch
    # boundary is at most floor(steps/2) steps ahead, positioning the
    # rollover inside the timed region at any --steps >= 2. This
    # over-weights reshuffle cost vs true steady state (1 rollover per
    # steps_per_epoch=50 steps), i.e. it is conservative.
    steps_per_epoch = args.rows_per_gpu // args.batch_size
    burn_max = steps_per_epoch  # at most one epoch of positioning
    total_steps = args.warmup + burn_max + args.steps
    num_epochs = (total_steps + steps_per_epoch - 1) // steps_per_epoch + 1

    feature_columns = [f"f{i}" for i in range(args.num_cols)]
    use_bf16 = args.dtype == "bf16"
    ds = TorchShufflingDataset(
        filenames,
        num_epochs,
        num_trainers=world,
        batch_size=args.batch_size,
        rank=rank,
        drop_last=True,
        num_reducers=args.reducers_per_gpu * world,
        max_concurrent_epochs=args.max_concurrent_epochs,
        feature_columns=feature_columns,
        feature_types=(
            [torch.bfloat16] * args.num_cols if use_bf16 else None
        ),
        label_column="labels",
        feature_matrix=True,
        device=device,
        source_cache=args.source_cache,
        output=args.loader_output,
    )

    # Default train step: the fused chain-kernel path (fwd+loss+bwd in two
    # hand-written MFMA kernels + split-K wgrads — measured faster than
    # eager autocast; RSDL_FUSED_STEP=0 reverts). At world>1 the model is
    # NOT DDP-wrapped on the fused path: gradients land in views of one
    # flat buffer and are all-reduced with a single collective (DDP's
    # autograd-hook bucketing never fires on a manual backward).
    model = TabularMLP(args.num_cols).to(device)
    use_fused = (
        os.environ.get("RSDL_FUSED_STEP", "1") == "1"
        and device.type == "cuda"
        and args.dtype == "bf16"
        and args.num_cols == 100
    )
    flat_grad = None
    grad_regions = {}
    if use_fused:
        if world > 1:
            params = list(model.parameters())
            # DDP would broadcast at wrap time; do it explicitly here.
            for p in params:
                torch.distributed.broadcast(p.data, src=0)
            flat_grad, grad_regions = flat_grad_views(model, device)
            model._rsdl_flat_grads = True
    elif world > 1:
        model = torch.nn.parallel.DistributedDataParallel(model)
    try:
        # Single fused multi-tensor update kernel (falls back where the
        # fused path is unavailable, e.g. CPU).
        opt = torch.optim.SGD(
            model.parameters(), lr=1e-3, momentum=0.9, fused=True
        )
    except (RuntimeError, TypeError, ValueError):
        opt = torch.optim.SGD(model.parameters(), lr=1e-3, momentum=0.9)
    loss_fn = torch.nn.MSELoss()

    def batches():
        for epoch in range(num_epochs):
            ds.set_epoch(epoch)
            for item in ds:
                yield epoch, item

    it = batches()
    cur_epoch = [0]
    is_cuda = device.type == "cuda"

    def sync():
        if is_cuda:
            torch.cuda.synchronize(device)

    def barrier():
        if world > 1:
            torch.distributed.barrier()

    import contextlib

    def amp():
        if use_bf16:
            return torch.autocast(
                device_type=device.type, dtype=torch.bfloat16
            )
        return contextlib.nullcontext()

    if use_fused:
        from ray_shuffling_data_loader_amd.models.fused_step import (
            fused_step,
        )

    # RSDL_OVERLAP_ALLREDUCE=1 (flat-grad DP only): all-reduce each grad
    # region async as fused_step reports it ready — the bias+head slice
    # overlaps all three wgrad kernels, dW1's overlaps dW2+dW3's.
    overlap_ar = (
        flat_grad is not None
        and os.environ.get("RSDL_OVERLAP_ALLREDUCE", "0") == "1"
    )
    ar_works = []

    def _overlap_hook(stage):
        s, e = grad_regions[stage]
        sl = flat_grad[s:e]
        sl.div_(world)
        ar_works.append(
            torch.distributed.all_reduce(sl, async_op=True)
        )

    def one_step():
        t_wait0 = time.perf_counter()
        cur_epoch[0], (data, target) = next(it)
        wait = time.perf_counter() - t_wait0
        x = data[0]
        if x.device != device:
            x = x.to(device, non_blocking=True)
            target = target.to(device, non_blocking=True)
        if use_fused:
            if overlap_ar:
                ar_works.clear()
                fused_step(model, x, target, grad_hook=_overlap_hook)
                for w in ar_works:
                    w.wait()
            else:
                fused_step(model, x, target)
                if flat_grad is not None:
                    flat_grad.div_(world)
                    torch.distributed.all_reduce(flat_grad)
            opt.step()
            return wait
        opt.zero_grad(set_to_none=True)
        with amp():
            out = model(x)
            loss = loss_fn(out.float(), target)
        loss.backward()
        opt.step()
        return wait

    # RSDL_HIP_GRAPH=1 (world 1, CUDA, fixed batch shape): capture the
    # whole train step (fwd+loss+bwd+opt) in a hipGraph after warmup and
    # replay it per step, copying each batch into static buffers first.
    # Removes per-step launch overhead (~25 launches -> 1 replay + 2
    # copies). Opt-in: at N>1 DDP comm hooks are not captured here, and a
    # graphed N=1 would skew the weak-scaling curve vs ungraphed N>1.
    use_graph = (
        os.environ.get("RSDL_HIP_GRAPH", "0") == "1"
        and is_cuda
        and world == 1
    )
    graph_state = {}

    def capture_graph(x_example, t_example):
        sx = torch.empty_like(x_example)
        st = torch.empty_like(t_example)

        def step_body():
            if use_fused:
                fused_step(model, sx, st)
            else:
                opt.zero_grad(set_to_none=False)
                with amp():
                    out = model(sx)
                    loss = loss_fn(out.float(), st)
                loss.backward()
            opt.step()
            if not use_fused:
                for p in model.parameters():
                    p.grad.zero_()

        side = torch.cuda.Stream(device)
        side.wait_stream(torch.cuda.current_stream(device))
        with torch.cuda.stream(side):
            for _ in range(3):  # allocator/optimizer state warmup
                step_body()
        torch.cuda.current_stream(device).wait_stream(side)
        g = torch.cuda.CUDAGraph()
        with torch.cuda.graph(g, capture_error_mode="thread_local"):
            step_body()
        graph_state.update(g=g, sx=sx, st=st)

    def one_step_graphed():
        t_wait0 = time.perf_counter()
        cur_epoch[0], (data, target) = next(it)
        wait = time.perf_counter() - t_wait0
        x = data[0]
        if "g" not in graph_state:
            capture_graph(x, target)
        graph_state["sx"].copy_(x, non_blocking=True)
        graph_state["st"].copy_(target, non_blocking=True)
        graph_state["g"].replay()
        return wait

    if use_graph:
        one_step = one_step_graphed
    elif is_cuda and os.environ.get("RSDL_HIPRI_STEP", "0") == "1":
        # A/B knob: run the train step on a HIGH-priority stream so its
        # kernels win CU arbitration against the side-stream reshuffle on
        # rollover-adjacent steps (the ~15% gap between the 20-step driver
        # window and steady state — profiles/PERF.md). Dequeued batches
        # record_stream() against the current stream at dequeue time, so
        # dequeuing inside the context keeps allocator safety.
        _hipri = torch.cuda.Stream(device=device, priority=-1)
        _base_step = one_step

        def one_step_hipri():
            with torch.cuda.stream(_hipri):
                return _base_step()

        one_step = one_step_hipri

    # Hang watchdog: if the whole job wedges (e.g. an RCCL collective
    # deadlock at N>1), dump every thread's stack to stderr so the failure
    # is diagnosable from one run. Armed around the measured region.
    import faulthandler

    watchdog_s = float(os.environ.get("RSDL_WATCHDOG_S", "300"))
    if watchdog_s > 0:
        faulthandler.dump_traceback_later(
            watchdog_s, repeat=True, exit=False
        )

    def progress(stage, i):
        if os.environ.get("RSDL_VERBOSE") == "1":
            print(
                f"[bench] rank {rank} {stage} step {i} "
                f"epoch {cur_epoch[0]}",
                file=sys.stderr,
                flush=True,
            )

    # Track position within the current epoch by observing actual epoch
    # transitions (robust to the +-1 batch-count variation of the
    # distributed binomial row split, where nominal arithmetic drifts).
    last_epoch = [0]
    in_epoch = [0]

    def step_tracked():
        w = one_step()
        if cur_epoch[0] != last_epoch[0]:
            last_epoch[0] = cur_epoch[0]
            in_epoch[0] = 1
        else:
            in_epoch[0] += 1
        return w

    try:
        for i in range(args.warmup):
            step_tracked()
            progress("warmup", i)
        # Position the epoch boundary inside the timed window (see above).
        burned = 0
        target_pos = steps_per_epoch - max(1, args.steps // 2)
        while in_epoch[0] < target_pos and burned < burn_max:
            step_tracked()
            burned += 1
        progress("burn", burned)
    except Exception:
        print(
            f"[bench] rank {rank} FAILED in warmup/burn at epoch "
            f"{cur_epoch[0]}",
            file=sys.stderr,
            flush=True,
        )
        raise

    sync()
    barrier()
    sync()
    prof = None
    if os.environ.get("RSDL_TRACE_FIRST") == "1" and rank == 0:
        prof = torch.profiler.profile(
            activities=[
                torch.profiler.ProfilerActivity.CPU,
                torch.profiler.ProfilerActivity.CUDA,
            ],
            with_stack=False,
        )
        prof.__enter__()
    t0 = time.perf_counter()
    waits = []
    rollovers = 0
    try:
        for i in range(args.steps):
            e_before = cur_epoch[0]
            waits.append(step_tracked())
            if cur_epoch[0] != e_before:
                rollovers += 1
            if prof is not None and i == 2:
                prof.__exit__(None, None, None)
                print(
                    prof.key_averages().table(
                        sort_by="self_cpu_time_total", row_limit=18
                    ),
                    flush=True,
                )
                prof = None
    except Exception:
        print(
            f"[bench] rank {rank} FAILED in timed loop at step "
            f"{len(waits)} epoch {cur_epoch[0]}",
            file=sys.stderr,
            flush=True,
        )
        raise
    sync()
    barrier()
    sync()
    elapsed = time.perf_counter() - t0
    if watchdog_s > 0:
        faulthandler.cancel_dump_traceback_later()
    if os.environ.get("RSDL_DEBUG_WAITS") == "1" and rank == 0:
        top = sorted(enumerate(waits), key=lambda kv: -kv[1])[:5]
        print(
            "[debug] top waits (step_idx, ms):",
            [(i, round(w * 1e3, 2)) for i, w in top],
            flush=True,
        )

    # MAX over ranks (slowest rank defines job time).
    if world > 1:
        t = torch.tensor([elapsed], dtype=torch.float64)
        if torch.distributed.get_backend() == "nccl":
            t = t.to(device)
        torch.distributed.all_reduce(
            t, op=torch.distributed.ReduceOp.MAX
        )
        elapsed = float(t.cpu().item())

    # Drain the remaining epochs so engine threads/collectives finish clean.
    drained = 0
    try:
        for _ in it:
            drained += 1
    except Exception:
        pass

    if os.environ.get("RSDL_LOG_MEM") == "1" and is_cuda and rank == 0:
        # stderr: stdout carries exactly ONE JSON line (driver contract)
        print(
            f"[mem] allocated={torch.cuda.memory_allocated()/2**30:.2f}GiB "
            f"max_allocated={torch.cuda.max_memory_allocated()/2**30:.2f}GiB "
            f"reserved={torch.cuda.memory_reserved()/2**30:.2f}GiB",
            flush=True,
            file=sys.stderr,
        )

    rows_per_sec = n_gpus * args.batch_size * args.steps / elapsed
    p50_wait_ms = statistics.median(waits) * 1000 if waits else None
    if rank == 0:
        result = {
            "metric": "shuffled_rows_per_sec",
            "value": rows_per_sec,
            "unit": "rows/s",
            "n_gpus": n_gpus,
            "steps": args.steps,
            "warmup": args.warmup,
            "ms_per_step": elapsed / args.steps * 1000,
            "higher_is_better": True,
            "scaling": "weak",
            "vs_baseline": None,
            "dtype": args.dtype,
            "data": (
                "synthetic parquet (float64 source cols), generated locally; "
                f"cached {'in HBM' if args.source_cache in ('auto', 'device') else args.source_cache} "
                "after first read; full per-epoch reshuffle "
                "(assignment+all-to-all+permute) every epoch"
            ),
            "config": {
                "model": "shuffling-data-loader + TabularMLP(100-512-256-128-1)",
                "global_batch": args.batch_size * n_gpus,
                "rows_per_gpu": args.rows_per_gpu,
                "num_cols": args.num_cols,
                "batch_size_per_rank": args.batch_size,
                "num_reducers": args.reducers_per_gpu * world,
                "max_concurrent_epochs": args.max_concurrent_epochs,
                "parallelism": f"dp{n_gpus}",
                # Proof the timed region crossed >= 1 epoch boundary, i.e.
                # the number includes a full per-epoch reshuffle: untimed
                # positioning steps burned after warmup, and the count of
                # epoch rollovers observed inside the timed window.
                "burn_steps": burned,
                "epoch_rollovers_in_timed_window": rollovers,
                "p50_batch_wait_ms": p50_wait_ms,
                "mean_batch_wait_ms": (
                    sum(waits) / len(waits) * 1000 if waits else None
                ),
                "p95_batch_wait_ms": (
                    sorted(waits)[int(0.95 * len(waits))] * 1000
                    if waits
                    else None
                ),
                "max_batch_wait_ms": (
                    max(waits) * 1000 if waits else None
                ),
            },
        }
        # Self-describing A/B records: any non-default RSDL_* knob that
        # was set is named in the JSON (empty when absent = stock run).
        knobs = {
            k: v
            for k, v in sorted(os.environ.items())
            if k.startswith("RSDL_")
            and k
            not in ("RSDL_NCCL_DEBUG", "RSDL_WATCHDOG_S", "RSDL_TUNABLEOP")
        }
        if knobs:
            result["config"]["knobs"] = knobs
        print(json.dumps(result), flush=True)

    if world > 1:
        torch.distributed.destroy_process_group()

    # Driver hygiene: the SCALE tier runs N=1,2,4,8 back-to-back on one box;
    # each N generates its own shard dir (up to ~40 GB at N=8). When using
    # the DEFAULT data dir, each rank removes its own generated files so
    # runs don't accumulate on /tmp. An explicit --data-dir is preserved
    # (cached reuse).
    if args.data_dir is None:
        import shutil

        for i in my_indices:
            try:
                os.remove(filenames[i])
            except OSError:
                pass
        if rank == 0:
            shutil.rmtree(shard_dir, ignore_errors=True)


if __name__ == "__main__":
    main()
