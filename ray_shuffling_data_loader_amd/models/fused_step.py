"""Fused train step for the fixed TabularMLP(100-512-256-128-1) — the
DEFAULT bench step (RSDL_FUSED_STEP=0 reverts to eager autocast).

No autograd graph; the step IS the schedule (~12 launches):
  x swizzle (emits both x layouts) -> fwd chain (3 MFMA layers + head +
  MSE loss/grad; emits a1^T/a2^T wgrad fragments + relu-mask words) ->
  bwd chain (dgrad chain from mask bits; emits dz^T fragments + db/dW4
  partials) -> 3 fragment-major wgrad kernels -> slab-partial reduces ->
  optimizer.
Activations a1/a2 and gradients dz never exist in row-major form: the
producers emit the transposed fragment layout the wgrad kernel reads
(csrc/fwd_chain.hip, csrc/bwd_chain.hip, csrc/wgrad_frag.hip), and the
backward reads 1-bit relu masks instead of activation values.

Validated by tests/test_gpu_kernels.py (chain numerics + whole-step
parity vs eager autograd) and the lane-level CPU simulation in
tests/test_chain_sim.py.
"""

import os
from typing import Tuple

import torch

from ray_shuffling_data_loader_amd.models.mlp import TabularMLP

# pi16 emission layout (RSDL_PI16=1): the chain kernels write each MFMA
# half-wave's packed pairs directly as the transposed-fragment runs
# under a shared intra-chunk M-permutation, dropping the cross-lane
# exchange from both epilogues. dW is invariant (M is the contraction
# dim of every consumer); schedule evidence in profiles/r02. Tests may
# override via fused_step._PI16.
_PI16 = os.environ.get("RSDL_PI16", "0") == "1"


def _layers(model: TabularMLP):
    """The four Linear modules of the fixed architecture, in order."""
    lin = [m for m in model.modules() if isinstance(m, torch.nn.Linear)]
    assert len(lin) == 4, "fused_step requires the stock TabularMLP(100)"
    shapes = [(512, 100), (256, 512), (128, 256), (1, 128)]
    for m, (n, k) in zip(lin, shapes):
        assert m.weight.shape == (n, k), (m.weight.shape, (n, k))
    return lin


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
    # One multi-tensor cast-copy instead of four kernels.
    torch._foreach_copy_(
        [buf["W1p"][:, :100], buf["W2"], buf["W3"], buf["w4"]],
        [
            lin[0].weight.detach(),
            lin[1].weight.detach(),
            lin[2].weight.detach(),
            lin[3].weight.detach().view(-1),
        ],
    )
    return buf


def fused_step(
    model: TabularMLP, x: torch.Tensor, target: torch.Tensor,
    grad_hook=None,
) -> Tuple[torch.Tensor, None]:
    """One manual fwd+bwd: computes the MSE loss and POPULATES .grad on
    every parameter of ``model`` (fp32, ready for an optimizer step).
    ``x`` is the bf16 [M,100] feature batch; ``target`` is [M,1].
    Returns the (scalar fp32) loss.

    ``grad_hook`` (flat-grad DP mode): called as grads become ready —
    ``grad_hook("bias")`` once every bias grad AND the head weight grad
    are in their views (right after the backward chain, BEFORE the
    wgrad kernels), then ``grad_hook("w1"|"w2"|"w3")`` after each weight
    grad copy. Lets the caller overlap per-slice gradient collectives
    with the remaining wgrad kernels (bench RSDL_OVERLAP_ALLREDUCE)."""
    from ray_shuffling_data_loader_amd.ops.shuffle_ops import _load_hip

    hip = _load_hip()
    lin = _layers(model)
    M = x.shape[0]
    from ray_shuffling_data_loader_amd.ops.shuffle_ops import wgrad_frag

    buf = _weight_buffers(model, lin)
    b1, b2, b3, b4 = (m.bias.detach() for m in lin)
    # Forward chain: a1/a2 exist only TRANSPOSED (wgrad fragment-major)
    # plus 32-bit relu-mask words; loss + dy fold into the head epilogue.
    # The x swizzle emits BOTH x layouts (fwd fragments + wgrad x^T) in
    # one pass over x.
    mchunks = 2 * ((M + 31) // 32)
    xt = torch.empty(4 * mchunks * 512, dtype=torch.bfloat16,
                     device=x.device)
    a1t, mask1, a2t, mask2, a3, out, dyb, loss_part = hip.fwd_chain_bf16(
        x, buf["W1p"], b1, buf["W2"], b2, buf["W3"], b3, buf["w4"], b4,
        target=target, xt_out=xt, pi16=_PI16,
    )
    loss = loss_part.sum() / M
    # Backward chain: consumes the masks (never the activations), emits
    # dz^T fragments; dW4/db4 partials fold into its seed loop.
    dz1t, dz2t, dz3t, db1, db2, db3, db4, dw4 = hip.bwd_chain_bf16(
        dyb, a3, mask1, mask2, buf["w4"], buf["W3"], buf["W2"], pi16=_PI16
    )
    # The DP bench pre-creates .grad as views of one flat buffer (so the
    # gradient all-reduce is a single collective) and marks the model;
    # only then do we COPY into them. Otherwise assign the fresh tensors
    # (copying would add 8 small kernels per step at N=1).
    flat_mode = getattr(model, "_rsdl_flat_grads", False)
    if flat_mode and grad_hook is not None:
        # Bias + head grads are ready NOW — copy and signal so their
        # collective overlaps the wgrad kernels below.
        for m, gb in zip(lin, (db1, db2, db3, db4)):
            m.bias.grad.copy_(gb.reshape(m.bias.shape))
        lin[3].weight.grad.copy_(dw4)
        grad_hook("bias")
    # Weight grads: fragment-major MFMA wgrad kernel (csrc/wgrad_frag.hip)
    # reading the transposed fragments the producers emitted.
    dw1 = wgrad_frag(dz1t, xt, 512, 128, mchunks)[:, :100].contiguous()
    if flat_mode and grad_hook is not None:
        lin[0].weight.grad.copy_(dw1)
        grad_hook("w1")
    dw2 = wgrad_frag(dz2t, a1t, 256, 512, mchunks)
    if flat_mode and grad_hook is not None:
        lin[1].weight.grad.copy_(dw2)
        grad_hook("w2")
    dw3 = wgrad_frag(dz3t, a2t, 128, 256, mchunks)
    if flat_mode and grad_hook is not None:
        lin[2].weight.grad.copy_(dw3)
        grad_hook("w3")
        return loss
    grads = [
        (dw1, db1),
        (dw2, db2),
        (dw3, db3),
        (dw4, db4),
    ]
    for m, (gw, gb) in zip(lin, grads):
        gb = gb.reshape(m.bias.shape)
        if flat_mode:
            m.weight.grad.copy_(gw)
            m.bias.grad.copy_(gb)
        else:
            m.weight.grad = gw.to(m.weight.dtype)
            m.bias.grad = gb.to(m.bias.dtype)
    return loss
