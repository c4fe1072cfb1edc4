"""Trainer models for the shuffling-loader benchmarks/examples.

The reference's Horovod example defines an MNIST-ish conv ``Net``
(reference: examples/horovod/ray_torch_shuffle.py:124-140) but never runs it
— the train step is ``time.sleep(mock_train_step_time)`` with the real
fwd/bwd commented out (ray_torch_shuffle.py:209-218). Here the benchmark
model is a real MLP regression head shaped for the tabular batches the
loader yields, and the step executes a genuine forward+backward+optimizer
update on the GPU.
"""

import os

import torch
import torch.nn as nn

# The hand-written MFMA wgrad kernel (csrc/wgrad_kernel.hip) is numerically
# verified and reaches 0.15-0.48 ms per layer shape, but the chunked-bmm
# composite is still faster end-to-end on gfx950 (0.70 vs 1.09 ms across
# the three layers incl. bias sums — profiles/PERF.md); default to bmm,
# opt in with RSDL_WGRAD_KERNEL=1.
_USE_WGRAD_KERNEL = os.environ.get("RSDL_WGRAD_KERNEL", "0") == "1"

# Fused ReLU-backward + bias-grad kernel (csrc/relu_bwd.hip): one HBM pass
# for dx = dy * (y > 0) AND db = dx.sum(0), replacing threshold_backward +
# a separate column reduce. bf16 GPU tensors with power-of-two width only;
# anything else falls back to the torch ops below.
_USE_FUSED_RELU_BWD = os.environ.get("RSDL_FUSED_RELU_BWD", "1") == "1"


def _relu_bwd_bias(dy, y):
    """Returns (masked dy, bias grad) via the fused kernel, or
    (masked dy, None) on the torch fallback path."""
    n = dy.shape[1]
    if (
        _USE_FUSED_RELU_BWD
        and dy.is_cuda
        and dy.dtype == torch.bfloat16
        and y.dtype == torch.bfloat16
        and 8 <= n <= 2048
        and (n & (n - 1)) == 0
    ):
        from ray_shuffling_data_loader_amd.ops import shuffle_ops

        hip = shuffle_ops._load_hip()
        dx, db = hip.relu_bwd_bias(dy.contiguous(), y.contiguous())
        return dx, db.to(dy.dtype)
    return torch.ops.aten.threshold_backward(dy, y, 0), None


def _wgrad_bf16_kernel(dy, x, with_bias):
    """Fused MFMA split-M wgrad (+bias grad) — csrc/wgrad_kernel.hip."""
    from ray_shuffling_data_loader_amd.ops.shuffle_ops import wgrad

    dw, db = wgrad(dy, x, with_bias)
    return dw.to(dy.dtype), (db.to(dy.dtype) if db is not None else None)


_WGRAD_CHUNK_TARGET = int(os.environ.get("RSDL_BMM_CHUNKS", "50"))


def _wgrad_chunks(m: int, target: int = None) -> int:
    """Largest divisor of m that is <= target (split-K chunk count).
    Tune with RSDL_BMM_CHUNKS: chunk count x output tiles must exceed the
    chip's 256 CUs by a healthy margin (profiles/PERF.md)."""
    if target is None:
        target = _WGRAD_CHUNK_TARGET
    for c in range(min(target, m), 0, -1):
        if m % c == 0:
            return c
    return 1


class _ChunkedLinearFn(torch.autograd.Function):
    """Linear with a split-K weight gradient.

    hipBLASLt's single-GEMM wgrad for tall-K shapes (dW[N,K] = dy^T @ x with
    K = batch = 250k) runs ~10x off the memory roofline on gfx950 (0.5 ms,
    0.6 TB/s, even TunableOp-tuned — profiles/PERF.md). Computing it as a
    batched GEMM over K-chunks + sum forces proper split-K parallelism:
    0.115 ms (2.7 TB/s) for the same shape. This was the dominant cost of
    the benchmark train step (3 wgrads ~= 1.9 of 2.9 ms)."""

    @staticmethod
    def forward(ctx, x, weight, bias):
        ctx.save_for_backward(x, weight)
        ctx.has_bias = bias is not None
        return torch.nn.functional.linear(x, weight, bias)

    @staticmethod
    def backward(ctx, dy):
        x, weight = ctx.saved_tensors
        dy = dy.contiguous()
        m = x.shape[0]
        n = dy.shape[1]
        if n == 1:
            # Scalar-output head: the "GEMMs" degenerate. dgrad is an outer
            # product dy[M,1] @ W[1,K] == broadcast multiply (hipBLASLt ran
            # it at 132 us/step vs ~15 us roofline, profiles/PERF.md), and
            # wgrad is a column-weighted reduce.
            dx = dy * weight if ctx.needs_input_grad[0] else None
            dw = (dy * x).sum(0, keepdim=True)
            db = dy.sum(0) if ctx.has_bias else None
            return dx, dw, db
        dx = dy @ weight if ctx.needs_input_grad[0] else None
        # Chunking only pays for wide-output, tall-K wgrads; for skinny
        # outputs hipBLASLt's plain mm is fine and the batched kernel is
        # pathological (11.7 ms for N=1 vs 0.13).
        c = (
            _wgrad_chunks(m)
            if min(n, x.shape[1]) >= 32 and m >= 1 << 16
            else 1
        )
        if c > 1:
            dw = (
                torch.bmm(
                    dy.view(c, m // c, n).transpose(1, 2),
                    x.view(c, m // c, x.shape[1]),
                )
                .sum(0)
            )
        else:
            dw = dy.t() @ x
        db = dy.sum(0) if ctx.has_bias else None
        return dx, dw, db


class ChunkedLinear(nn.Linear):
    """nn.Linear drop-in using the split-K chunked weight gradient."""

    def forward(self, x):
        if x.is_cuda and x.dim() == 2:
            w, b = self.weight, self.bias
            # Match autocast semantics of nn.Linear.
            if torch.is_autocast_enabled():
                adt = torch.get_autocast_dtype("cuda")
                x = x.to(adt)
                w = w.to(adt)
                b = b.to(adt) if b is not None else None
            return _ChunkedLinearFn.apply(x, w, b)
        return super().forward(x)


class _LinearReLUFn(torch.autograd.Function):
    """Linear + bias + ReLU with the GEMM-epilogue fusion
    (aten._addmm_activation -> hipBLASLt ReLU epilogue: no separate
    bias-add/relu elementwise pass over the [M, N] activation) and the
    split-K chunked weight gradient. The ReLU mask for backward is derived
    from the saved OUTPUT (y > 0), so no extra mask tensor."""

    @staticmethod
    def forward(ctx, x, weight, bias):
        y = torch.ops.aten._addmm_activation(
            bias, x, weight.t(), use_gelu=False
        )
        ctx.save_for_backward(x, weight, y)
        return y

    @staticmethod
    def backward(ctx, dy):
        x, weight, y = ctx.saved_tensors
        # relu' (+ fused bias grad when the HIP kernel applies).
        dy, db = _relu_bwd_bias(dy, y)
        # First layer: input is the loader's feature batch (no grad) —
        # skip the [M,N]x[N,K] dgrad GEMM entirely.
        dx = dy @ weight if ctx.needs_input_grad[0] else None
        m = x.shape[0]
        if (
            _USE_WGRAD_KERNEL
            and dy.is_cuda
            and dy.dtype == torch.bfloat16
            and x.dtype == torch.bfloat16
            and m >= 1 << 16
        ):
            dw, db_w = _wgrad_bf16_kernel(dy, x, with_bias=db is None)
            return dx, dw, db if db is not None else db_w
        c = (
            _wgrad_chunks(m)
            if min(dy.shape[1], x.shape[1]) >= 32 and m >= 1 << 16
            else 1
        )
        if c > 1:
            dw = (
                torch.bmm(
                    dy.view(c, m // c, dy.shape[1]).transpose(1, 2),
                    x.view(c, m // c, x.shape[1]),
                )
                .sum(0)
            )
        else:
            dw = dy.t() @ x
        return dx, dw, db if db is not None else dy.sum(0)


class LinearReLU(nn.Linear):
    """Fused Linear+ReLU layer (GEMM epilogue ReLU + chunked wgrad)."""

    def forward(self, x):
        if x.is_cuda and x.dim() == 2 and self.bias is not None:
            w, b = self.weight, self.bias
            if torch.is_autocast_enabled():
                adt = torch.get_autocast_dtype("cuda")
                x = x.to(adt)
                w = w.to(adt)
                b = b.to(adt)
            return _LinearReLUFn.apply(x, w, b)
        return torch.relu(super().forward(x))


class TabularMLP(nn.Module):
    """MLP over a [B, num_features] float batch -> scalar regression."""

    def __init__(
        self,
        num_features: int = 100,
        hidden: int = 512,
        depth: int = 3,
        linear_cls=ChunkedLinear,
    ):
        super().__init__()
        layers = []
        d = num_features
        for i in range(depth):
            h = hidden // (2**i)
            if linear_cls is ChunkedLinear:
                layers.append(LinearReLU(d, h))
            else:
                layers += [linear_cls(d, h), nn.ReLU()]
            d = h
        layers.append(linear_cls(d, 1))
        self.net = nn.Sequential(*layers)

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        return self.net(x)


class EmbeddingMLP(nn.Module):
    """Model shaped for the reference DATA_SPEC batches: 19 int64
    categorical columns -> embeddings, concatenated -> MLP
    (the realistic counterpart of the reference example's unused Net)."""

    def __init__(self, cardinalities, embed_dim: int = 16, hidden: int = 256):
        super().__init__()
        self.embeddings = nn.ModuleList(
            [
                nn.Embedding(int(c), embed_dim)
                for c in cardinalities
            ]
        )
        d = embed_dim * len(cardinalities)
        self.mlp = nn.Sequential(
            nn.Linear(d, hidden),
            nn.ReLU(),
            nn.Linear(hidden, 1),
        )

    def forward(self, cat_feats) -> torch.Tensor:
        embs = [
            emb(x.clamp_min(0).squeeze(-1) % emb.num_embeddings)
            for emb, x in zip(self.embeddings, cat_feats)
        ]
        return self.mlp(torch.cat(embs, dim=1))
