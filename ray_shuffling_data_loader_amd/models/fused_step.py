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
    wb = [(m.weight.detach().bfloat16(), m.bias.detach()) for m in lin]
    (W1, b1), (W2, b2), (W3, b3), (W4, b4) = wb
    a1, a2, a3, out = hip.fwd_chain_bf16(
        x, W1, b1, W2, b2, W3, b3, W4.flatten(), b4
    )
    diff = out.float() - target.float().reshape(-1, 1)
    loss = diff.square().mean()
    dy = (2.0 / (M * 1.0)) * diff  # d(mean((out-t)^2))/d out
    dyb = dy.bfloat16().contiguous()
    dz1, dz2, dz3, db1, db2, db3, db4 = hip.bwd_chain_bf16(
        dyb, a1, a2, a3, W4.flatten(), W3, W2
    )
    grads = [
        (_wgrad_bmm(dz1, x).float(), db1),
        (_wgrad_bmm(dz2, a1).float(), db2),
        (_wgrad_bmm(dz3, a2).float(), db3),
        ((dyb.t().float() @ a3.float()), db4),
    ]
    for m, (gw, gb) in zip(lin, grads):
        m.weight.grad = gw.to(m.weight.dtype)
        m.bias.grad = gb.to(m.bias.dtype).reshape(m.bias.shape)
    return loss
