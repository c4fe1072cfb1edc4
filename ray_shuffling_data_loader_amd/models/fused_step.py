"""EXPERIMENTAL round-2 fused train step (docs/MEGAKERNEL_PLAN.md).

Manual forward+backward for the fixed TabularMLP(100-512-256-128-1)
architecture using the fused chain kernels (csrc/fwd_chain.hip,
csrc/bwd_chain.hip) plus the existing chunked-bmm split-K weight
gradients. No autograd graph: the step IS the schedule —
2 chain kernels + 4 wgrad GEMM groups + the caller's optimizer.

NOT wired into any default path and NOT yet GPU-validated (written after
round-1's GPU budget was spent); exercised by the RSDL_EXPERIMENTAL=1
GPU test. Round 2: validate numerics, A/B in bench.py behind
RSDL_FUSED_STEP=1, then consider hipGraph capture (the step shrinks to
~10 launches).
"""

from typing import Tuple

import torch

from ray_shuffling_data_loader_amd.models.mlp import TabularMLP, _wgrad_chunks


def _layers(model: TabularMLP):
    """The four Linear modules of the fixed architecture, in order."""
    lin = [m for m in model.modules() if isinstance(m, torch.nn.Linear)]
    assert len(lin) == 4, "fused_step requires the stock TabularMLP(100)"
    shapes = [(512, 100), (256, 512), (128, 256), (1, 128)]
    for m, (n, k) in zip(lin, shapes):
        assert m.weight.shape == (n, k), (m.weight.shape, (n, k))
    return lin


def _wgrad_bmm(dz: torch.Tensor, src: torch.Tensor) -> torch.Tensor:
    """dW = dz^T @ src via the split-K chunked bmm (models/mlp.py)."""
    m = dz.shape[0]
    c = _wgrad_chunks(m) if m >= 1 << 16 else 1
    if c > 1:
        return (
            torch.bmm(
                dz.view(c, m // c, dz.shape[1]).transpose(1, 2),
                src.view(c, m // c, src.shape[1]),
            )
            .sum(0)
        )
    return dz.t() @ src


def _weight_buffers(model: TabularMLP, lin):
    """Persistent bf16 weight staging for the chain kernels: W1 pre-padded
    to [512,112], W2/W3/w4 in model layout, W2T/W3T pre-transposed for the
    dgrad B fragments. Allocated once; refreshed per step with plain
    cast-copies (the params change every optimizer step)."""
    buf = getattr(model, "_fused_buf", None)
    dev = lin[0].weight.device
    if buf is None or buf["W1p"].device != dev:
        bf = dict(dtype=torch.bfloat16, device=dev)
        buf = {
            "W1p": torch.zeros(512, 112, **bf),
            "W2": torch.empty(256, 512, **bf),
            "W3": torch.empty(128, 256, **bf),
            "w4": torch.empty(128, **bf),
        }
        model._fused_buf = buf
    buf["W1p"][:, :100].copy_(lin[0].weight.detach())
    buf["W2"].copy_(lin[1].weight.detach())
    buf["W3"].copy_(lin[2].weight.detach())
    buf["w4"].copy_(lin[3].weight.detach().view(-1))
    return buf


def fused_step(
    model: TabularMLP, x: torch.Tensor, target: torch.Tensor
) -> Tuple[torch.Tensor, None]:
    """One manual fwd+bwd: computes the MSE loss and POPULATES .grad on
    every parameter of ``model`` (fp32, ready for an optimizer step).
    ``x`` is the bf16 [M,100] feature batch; ``target`` is [M,1].
    Returns the (scalar fp32) loss."""
    from ray_shuffling_data_loader_amd.ops.shuffle_ops import _load_hip

    hip = _load_hip()
    lin = _layers(model)
    M = x.shape[0]
    buf = _weight_buffers(model, lin)
    b1, b2, b3, b4 = (m.bias.detach() for m in lin)
    # Loss + dy are fused into the forward kernel's head epilogue.
    a1, a2, a3, out, dyb, loss_part = hip.fwd_chain_bf16(
        x, buf["W1p"], b1, buf["W2"], b2, buf["W3"], b3, buf["w4"], b4,
        target=target,
    )
    loss = loss_part.sum() / M
    # dW4/db4 partials are folded into the backward kernel's seed loop.
    dz1, dz2, dz3, db1, db2, db3, db4, dw4 = hip.bwd_chain_bf16(
        dyb, a1, a2, a3, buf["w4"], buf["W3"], buf["W2"]
    )
    grads = [
        (_wgrad_bmm(dz1, x), db1),
        (_wgrad_bmm(dz2, a1), db2),
        (_wgrad_bmm(dz3, a2), db3),
        (dw4, db4),
    ]
    for m, (gw, gb) in zip(lin, grads):
        # Copy into pre-existing .grad buffers when present (the DP bench
        # pre-creates them as views of one flat buffer so the gradient
        # all-reduce is a single collective); otherwise assign.
        if m.weight.grad is not None and m.weight.grad.shape == gw.shape:
            m.weight.grad.copy_(gw)
        else:
            m.weight.grad = gw.to(m.weight.dtype)
        gb = gb.reshape(m.bias.shape)
        if m.bias.grad is not None:
            m.bias.grad.copy_(gb)
        else:
            m.bias.grad = gb.to(m.bias.dtype)
    return loss
