"""Data-parallel wiring of the fused train step on CPU (gloo, world 2).

The bench's N>1 path is: broadcast params from rank 0, pre-create .grad
as views of one flat buffer (model marked `_rsdl_flat_grads`), run
fused_step (copies grads into the views), div by world, ONE all_reduce
of the flat buffer, optimizer step. This test executes exactly that
wiring with the torch FakeHip mirror of the chain kernels and asserts
both ranks end with identical parameters equal to a single-process
reference computing the averaged gradient.
"""

import os
import sys

import pytest
import torch


def _dp_worker(rank, world, port, result_q):
    try:
        sys.path.insert(
            0, os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
        )
        sys.path.insert(0, os.path.dirname(os.path.abspath(__file__)))
        import torch.distributed as dist

        from _fake_hip import FakeHip
        from ray_shuffling_data_loader_amd.models import fused_step as fs
        from ray_shuffling_data_loader_amd.models.mlp import TabularMLP
        import ray_shuffling_data_loader_amd.ops.shuffle_ops as so

        fs._load_hip = lambda: FakeHip
        so._load_hip = lambda: FakeHip

        os.environ["MASTER_ADDR"] = "127.0.0.1"
        os.environ["MASTER_PORT"] = str(port)
        dist.init_process_group("gloo", rank=rank, world_size=world)

        torch.manual_seed(100 + rank)  # DIFFERENT init per rank on purpose
        model = TabularMLP(100)
        params = list(model.parameters())
        # bench wiring: broadcast, flat-grad views, mark the model
        for p in params:
            dist.broadcast(p.data, src=0)
        flat = torch.zeros(sum(p.numel() for p in params))
        off = 0
        for p in params:
            p.grad = flat[off : off + p.numel()].view_as(p)
            off += p.numel()
        model._rsdl_flat_grads = True
        opt = torch.optim.SGD(model.parameters(), lr=1e-2, momentum=0.9)

        # per-rank batch (deterministic per rank)
        g = torch.Generator().manual_seed(7 + rank)
        x = torch.randn(512, 100, generator=g).bfloat16()
        t = torch.randn(512, 1, generator=g)

        fs.fused_step(model, x, t)
        flat.div_(world)
        dist.all_reduce(flat)
        opt.step()

        # plain numpy payloads: avoids torch's shared-memory tensor
        # reducers for the mp.Queue transport
        state = {
            n: p.detach().numpy().copy()
            for n, p in model.named_parameters()
        }
        result_q.put((rank, state, flat.numpy().copy()))
        dist.destroy_process_group()
    except Exception as e:  # noqa: BLE001
        import traceback

        result_q.put((rank, "ERR", f"{e}\n{traceback.format_exc()}"))


def test_fused_dp_world2_gloo(mp_spawn_context):
    import socket

    with socket.socket() as s:
        s.bind(("127.0.0.1", 0))
        port = s.getsockname()[1]
    rq = mp_spawn_context.Queue()
    procs = [
        mp_spawn_context.Process(target=_dp_worker, args=(r, 2, port, rq))
        for r in range(2)
    ]
    for p in procs:
        p.start()
    results = {}
    for _ in range(2):
        rank, state, extra = rq.get(timeout=300)
        assert state != "ERR", extra
        results[rank] = (state, extra)
    for p in procs:
        p.join(timeout=60)
        assert p.exitcode == 0

    import numpy as np

    # Ranks must agree bit-for-bit after the allreduce + step.
    for n in results[0][0]:
        assert np.array_equal(results[0][0][n], results[1][0][n]), n
    assert np.array_equal(results[0][1], results[1][1])

    # Single-process reference: rank-0 init (broadcast source), averaged
    # gradient over both ranks' batches.
    sys.path.insert(0, os.path.dirname(os.path.abspath(__file__)))
    from _fake_hip import FakeHip
    from ray_shuffling_data_loader_amd.models import fused_step as fs
    from ray_shuffling_data_loader_amd.models.mlp import TabularMLP
    import ray_shuffling_data_loader_amd.ops.shuffle_ops as so

    orig = so._load_hip
    so._load_hip = lambda: FakeHip
    try:
        torch.manual_seed(100)  # rank 0 init
        ref = TabularMLP(100)
        grads = {n: torch.zeros_like(p) for n, p in ref.named_parameters()}
        for r in range(2):
            g = torch.Generator().manual_seed(7 + r)
            x = torch.randn(512, 100, generator=g).bfloat16()
            t = torch.randn(512, 1, generator=g)
            import copy

            clone = copy.deepcopy(ref)
            fs.fused_step(clone, x, t)
            for n, p in clone.named_parameters():
                grads[n] += p.grad / 2
        opt = torch.optim.SGD(ref.parameters(), lr=1e-2, momentum=0.9)
        for n, p in ref.named_parameters():
            p.grad = grads[n]
        opt.step()
        for n, p in ref.named_parameters():
            assert torch.allclose(
                p.detach(), torch.from_numpy(results[0][0][n]), atol=1e-6
            ), n
    finally:
        so._load_hip = orig


def _overlap_worker(rank, world, port, result_q):
    try:
        sys.path.insert(
            0, os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
        )
        sys.path.insert(0, os.path.dirname(os.path.abspath(__file__)))
        import copy

        import torch.distributed as dist

        from _fake_hip import FakeHip
        from ray_shuffling_data_loader_amd.models import fused_step as fs
        from ray_shuffling_data_loader_amd.models.mlp import TabularMLP
        import ray_shuffling_data_loader_amd.ops.shuffle_ops as so

        so._load_hip = lambda: FakeHip

        os.environ["MASTER_ADDR"] = "127.0.0.1"
        os.environ["MASTER_PORT"] = str(port)
        dist.init_process_group("gloo", rank=rank, world_size=world)

        torch.manual_seed(321)
        base = TabularMLP(100)
        g = torch.Generator().manual_seed(70 + rank)
        x = torch.randn(256, 100, generator=g).bfloat16()
        t = torch.randn(256, 1, generator=g)

        def flatten(model):
            # bench layout: weights first, then biases — each grad_hook
            # stage is one contiguous region
            weights = [p for n, p in model.named_parameters()
                       if n.endswith("weight")]
            biases = [p for n, p in model.named_parameters()
                      if n.endswith("bias")]
            ordered = weights + biases
            flat = torch.zeros(sum(p.numel() for p in ordered))
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
                "bias": (offs[3], off),
            }
            model._rsdl_flat_grads = True
            return flat, regions

        # Mode A: single collective after the step.
        ma = copy.deepcopy(base)
        fa, _ = flatten(ma)
        fs.fused_step(ma, x, t)
        fa.div_(world)
        dist.all_reduce(fa)

        # Mode B: per-region async collectives via grad_hook.
        mb = copy.deepcopy(base)
        fb, regions = flatten(mb)
        works = []

        def hook(stage):
            s, e = regions[stage]
            sl = fb[s:e]
            sl.div_(world)
            works.append(dist.all_reduce(sl, async_op=True))

        fs.fused_step(mb, x, t, grad_hook=hook)
        for w in works:
            w.wait()

        result_q.put(
            (rank, fa.numpy().copy(), fb.numpy().copy())
        )
        dist.destroy_process_group()
    except Exception as e:  # noqa: BLE001
        import traceback

        result_q.put((rank, "ERR", f"{e}\n{traceback.format_exc()}"))


def test_overlap_allreduce_matches_single_collective(mp_spawn_context):
    """The RSDL_OVERLAP_ALLREDUCE wiring (per-region async all-reduce
    driven by fused_step's grad_hook) must produce bit-identical flat
    gradients to the single post-step collective."""
    import socket

    import numpy as np

    with socket.socket() as s:
        s.bind(("127.0.0.1", 0))
        port = s.getsockname()[1]
    rq = mp_spawn_context.Queue()
    procs = [
        mp_spawn_context.Process(
            target=_overlap_worker, args=(r, 2, port, rq)
        )
        for r in range(2)
    ]
    for p in procs:
        p.start()
    results = {}
    for _ in range(2):
        rank, fa, fb = rq.get(timeout=300)
        assert not (isinstance(fa, str) and fa == "ERR"), fb
        results[rank] = (fa, fb)
    for p in procs:
        p.join(timeout=60)
        assert p.exitcode == 0
    for rank in (0, 1):
        fa, fb = results[rank]
        assert np.array_equal(fa, fb), rank
    assert np.array_equal(results[0][0], results[1][0])
