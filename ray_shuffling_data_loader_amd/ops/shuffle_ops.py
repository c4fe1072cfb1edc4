"""Shuffle hot ops: gather-permute, pack, unpack — HIP on GPU, torch/numpy on
CPU.

GPU tensors dispatch to the gfx950 HIP kernels in ``_rsdl_hip``
(csrc/shuffle_kernels.hip). If the extension is missing on a GPU box the ops
raise immediately — there is deliberately NO eager/PyTorch fallback on GPU,
so a silent-slow path can't masquerade as the native one. CPU tensors use
torch/numpy implementations, which double as the fp32 oracle for the GPU
numerics tests.

Reference ops replaced here: ``pd.concat`` + ``df.sample(frac=1)``
(reference shuffle.py:192-194) -> :func:`gather_rows`;
``convert_to_tensor`` column cast/pack (reference torch_dataset.py:204-236)
-> :func:`unpack_permute`.
"""

from typing import Dict, Optional

import os

import numpy as np
import torch

from ray_shuffling_data_loader_amd.utils.schema import (
    Schema,
    TORCH_TO_NUMPY_DTYPE,
)

_hip = None
_hip_err = None


def _load_hip():
    global _hip, _hip_err
    if _hip is None and _hip_err is None:
        try:
            from ray_shuffling_data_loader_amd import _rsdl_hip

            _hip = _rsdl_hip
        except ImportError as e:  # remember why, re-raise on GPU use
            _hip_err = e
    if _hip is None:
        raise RuntimeError(
            "ray_shuffling_data_loader_amd._rsdl_hip extension is not built "
            "but a GPU tensor was passed to a shuffle op. Build it with "
            "`python setup.py build_ext --inplace` (PYTORCH_ROCM_ARCH="
            f"gfx950). Original import error: {_hip_err}"
        )
    return _hip


_DT_CODE = None


def _dtype_code(dtype: torch.dtype) -> int:
    global _DT_CODE
    if _DT_CODE is None:
        hip = _load_hip()
        _DT_CODE = {
            torch.float32: hip.DT_F32,
            torch.float64: hip.DT_F64,
            torch.int32: hip.DT_I32,
            torch.int64: hip.DT_I64,
            torch.float16: hip.DT_F16,
            torch.bfloat16: hip.DT_BF16,
            torch.uint8: hip.DT_U8,
        }
    return _DT_CODE[dtype]


# ---------------------------------------------------------------------------
# gather_rows: out[i,:] = src[perm[i],:]  on a packed byte matrix.
# ---------------------------------------------------------------------------


def gather_rows(
    src: torch.Tensor,
    perm: torch.Tensor,
    out: Optional[torch.Tensor] = None,
) -> torch.Tensor:
    """Row gather of a 2-D contiguous tensor. The reducer-side full row
    permutation (reference shuffle.py:194) fused with the implicit concat."""
    if src.is_cuda:
        hip = _load_hip()
        if out is not None:
            hip.gather_rows_out(src, perm, out)
            return out[: perm.numel()]
        return hip.gather_rows(src, perm)
    res = torch.index_select(src, 0, perm.to(torch.long))
    if out is not None:
        out[: perm.numel()] = res
        return out[: perm.numel()]
    return res


# ---------------------------------------------------------------------------
# pack / unpack between column tensors and a packed row-major byte matrix.
# ---------------------------------------------------------------------------


def _np_strided_view(
    packed_np: np.ndarray, schema: Schema, name: str, numel: int, np_dt
) -> np.ndarray:
    """Writable strided numpy view of one column inside a packed row buffer."""
    n = packed_np.shape[0]
    stride = packed_np.shape[1]
    itemsize = np_dt.itemsize
    return np.ndarray(
        shape=(n, numel),
        dtype=np_dt,
        buffer=packed_np,
        offset=schema.offsets[name],
        strides=(stride, itemsize),
    )


def pack_columns(
    columns: Dict[str, torch.Tensor],
    schema: Schema,
    perm: Optional[torch.Tensor] = None,
    out: Optional[torch.Tensor] = None,
) -> torch.Tensor:
    """Interleave column tensors into packed rows (cast to the schema dtype
    per column). ``perm`` scatters row i of the input to row perm[i] of the
    output."""
    first = next(iter(columns.values()))
    n = first.shape[0]
    if first.is_cuda:
        hip = _load_hip()
        cols, offs, codes = [], [], []
        for spec in schema.columns:
            t = columns[spec.name]
            cols.append(t.contiguous())
            offs.append(schema.offsets[spec.name])
            codes.append(_dtype_code(spec.dtype))
        if perm is None:
            # LDS-tiled transpose pack: coalesced column reads + one
            # contiguous row-major store stream (optionally in place).
            return hip.pack_columns_tiled(
                cols, offs, codes, schema.row_stride, out
            )
        return hip.pack_columns(
            cols, offs, codes, schema.row_stride, perm
        )
    packed = (
        out
        if out is not None
        else torch.zeros(n, schema.row_stride, dtype=torch.uint8)
    )
    packed_np = packed.numpy()
    perm_np = perm.cpu().numpy() if perm is not None else None
    for spec in schema.columns:
        t = columns[spec.name].detach()
        np_dt = TORCH_TO_NUMPY_DTYPE[spec.dtype]
        view = _np_strided_view(packed_np, schema, spec.name, spec.numel, np_dt)
        src = t.cpu().numpy().reshape(n, spec.numel).astype(np_dt, copy=False)
        if perm_np is not None:
            view[perm_np] = src
        else:
            view[:] = src
    return packed


def unpack_permute(
    packed: torch.Tensor,
    schema: Schema,
    perm: Optional[torch.Tensor] = None,
    out_dtypes: Optional[Dict[str, torch.dtype]] = None,
) -> Dict[str, torch.Tensor]:
    """Fused gather-permute + per-column cast + pack into contiguous torch
    column tensors: out[name][i] = cast(packed[perm[i], off_name]).

    This single op replaces the reference's reduce-side
    concat+sample+convert_to_tensor chain (shuffle.py:192-194 +
    torch_dataset.py:204-236)."""
    out_dtypes = out_dtypes or {}
    n = perm.numel() if perm is not None else packed.shape[0]
    if packed.is_cuda:
        hip = _load_hip()
        outs, offs, codes = [], [], []
        result: Dict[str, torch.Tensor] = {}
        for spec in schema.columns:
            dst_dt = out_dtypes.get(spec.name, spec.dtype)
            shape = (n,) if spec.numel == 1 else (n, spec.numel)
            o = torch.empty(shape, dtype=dst_dt, device=packed.device)
            outs.append(o)
            offs.append(schema.offsets[spec.name])
            codes.append(_dtype_code(spec.dtype))
            result[spec.name] = o
        hip.unpack_permute(packed, perm, outs, offs, codes)
        return result
    packed_np = packed.numpy()
    perm_np = perm.cpu().numpy() if perm is not None else None
    result = {}
    for spec in schema.columns:
        np_dt = TORCH_TO_NUMPY_DTYPE[spec.dtype]
        view = _np_strided_view(packed_np, schema, spec.name, spec.numel, np_dt)
        arr = view[perm_np] if perm_np is not None else np.ascontiguousarray(
            view
        )
        t = torch.from_numpy(arr)
        dst_dt = out_dtypes.get(spec.name, spec.dtype)
        if dst_dt != spec.dtype:
            t = t.to(dst_dt)
        if spec.numel == 1:
            t = t.reshape(n)
        result[spec.name] = t
    return result


# ---------------------------------------------------------------------------
# Row partition by destination (map-side scatter; reference shuffle.py:156-161
# boolean-mask loop). v1: stable sort + native gather; the sort keys are tiny
# (uint8-range dest ids) so torch's radix sort is cheap next to the row move.
# ---------------------------------------------------------------------------


def partition_rows(
    packed: torch.Tensor,
    dest: torch.Tensor,
    num_dests: int,
):
    """Group packed rows by destination id. Returns (regrouped rows, counts
    per destination).

    GPU: fused histogram -> scan -> rank/scatter builds the gather
    permutation in two O(N) index passes (no radix sort); the row move is
    the roofline gather. Order within a destination is block-local (a full
    random permutation is applied downstream either way). CPU: stable
    argsort (order matches the reference's boolean-mask partition)."""
    if packed.is_cuda:
        hip = _load_hip()
        perm, counts = hip.partition_build_perm(dest, num_dests)
        grouped = gather_rows(packed, perm)
        return grouped, counts
    counts = torch.bincount(dest, minlength=num_dests)
    order = torch.argsort(dest, stable=True)
    grouped = gather_rows(packed, order)
    return grouped, counts


# ---------------------------------------------------------------------------
# MFMA split-M weight gradient (trainer-side hot op; csrc/wgrad_kernel.hip).
# ---------------------------------------------------------------------------


def wgrad(dy: torch.Tensor, x: torch.Tensor, with_bias: bool = True):
    """dW = dy^T @ x (fp32) and db = dy.sum(0) in one fused MFMA kernel
    (bf16 inputs, split-M atomic reduction). CPU / non-bf16 fallback uses
    plain matmul."""
    if dy.is_cuda and dy.dtype == torch.bfloat16 and x.dtype == torch.bfloat16:
        hip = _load_hip()
        dw, db = hip.wgrad_bf16(dy.contiguous(), x.contiguous(), with_bias)
        return dw, (db if with_bias else None)
    dw = dy.t().float() @ x.float()
    db = dy.float().sum(0) if with_bias else None
    return dw, db


# pi16 intra-chunk M-permutation: position p <-> row with bits 2 and 3
# of the 4-bit within-chunk index swapped (an involution). Under it each
# MFMA half-wave's own packed pairs are already contiguous 8-element
# runs, so the chain kernels' emission needs no cross-lane exchange; dW
# is invariant because M is the contraction dim of every consumer.
_PI16_IDX = [(p & 3) | ((p & 4) << 1) | ((p & 8) >> 1) for p in range(16)]


def t_frag_swizzle(t: torch.Tensor, pi16: bool = False) -> torch.Tensor:
    """[M, C] -> fragment-major layout of t^T: flat [C/32][Mp/16][2][32][8]
    (Mp = M padded to a 16-multiple with zero rows). This is the input
    layout of the fragment-major wgrad kernel (csrc/wgrad_frag.hip); the
    hot producers (bwd_chain dz^T, fwd_chain a^T) emit it directly — this
    torch implementation is the oracle/fallback. ``pi16`` applies the
    pi16 intra-chunk M-permutation (must match the producers' flag)."""
    m, c = t.shape
    mp = (m + 15) // 16 * 16
    if mp != m:
        t = torch.nn.functional.pad(t, (0, 0, 0, mp - m))
    v = t.reshape(mp // 16, 16, c)
    if pi16:
        v = v[:, _PI16_IDX]
    v = v.reshape(mp // 16, 2, 8, c // 32, 32)
    return v.permute(3, 0, 1, 4, 2).contiguous().reshape(-1)


_WGRAD_FRAG_CFG = {
    # (N, K) -> (nt_w, kt_w); see csrc/wgrad_frag.hip launch configs
    (512, 128): (2, 4),
    (256, 512): (1, 8),
    (128, 256): (1, 8),
}
# RSDL_WGRAD_SMALL_TILES=1: (1,4) tiles for the (1,8) shapes — 64 acc
# regs (3 waves/SIMD), and the depth-3 fenced loop when RSDL_WGRAD_SCHED
# is also on (csrc/wgrad_frag.hip launch configs).
_WGRAD_FRAG_SMALL = {
    (512, 128): (2, 4),
    (256, 512): (1, 4),
    (128, 256): (1, 4),
}


def wgrad_frag(at_frag: torch.Tensor, bt_frag: torch.Tensor, n: int,
               k: int, mchunks: int) -> torch.Tensor:
    """dW = dz^T @ src from pre-swizzled fragment inputs (fp32 [N,K])."""
    cfg = (
        _WGRAD_FRAG_SMALL
        if os.environ.get("RSDL_WGRAD_SMALL_TILES", "0") == "1"
        else _WGRAD_FRAG_CFG
    )
    nt_w, kt_w = cfg[(n, k)]
    hip = _load_hip()
    return hip.wgrad_frag_bf16(at_frag, bt_frag, n, k, mchunks, nt_w, kt_w)


def t_frag_unswizzle(
    flat: torch.Tensor, m: int, c: int, pi16: bool = False
) -> torch.Tensor:
    """Inverse of :func:`t_frag_swizzle`: fragment-major transposed flat
    tensor -> row-major [m, c] (pad rows dropped)."""
    mchunks = flat.numel() // (c * 16)
    v = flat.reshape(c // 32, mchunks, 2, 32, 8)
    v = v.permute(1, 2, 4, 0, 3).reshape(mchunks, 16, c)
    if pi16:
        v = v[:, _PI16_IDX]  # involution: same index recovers rows
    return v.reshape(mchunks * 16, c)[:m].contiguous()


def relu_mask_words(a: torch.Tensor) -> torch.Tensor:
    """[M, N] activations -> [ceil(M/32), N] int32 relu-mask words (bit i
    of word (t, n) = a[t*32+i, n] > 0; pad rows 0). Torch oracle for the
    mask layout the forward chain kernel emits."""
    m, n = a.shape
    mt = (m + 31) // 32
    pad = mt * 32 - m
    bits = a > 0
    if pad:
        bits = torch.cat(
            [bits, torch.zeros(pad, n, dtype=torch.bool, device=a.device)]
        )
    bits = bits.view(mt, 32, n).long()
    weights = (1 << torch.arange(32, device=a.device, dtype=torch.long))
    words = (bits * weights.view(1, 32, 1)).sum(1)
    return (words & 0xFFFFFFFF).to(torch.int32) if False else (
        words - ((words >> 31) & 1) * (1 << 32)
    ).to(torch.int32)
