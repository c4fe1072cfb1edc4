"""Exact torch mirror of the _rsdl_hip chain-kernel binding surface, used
to exercise fused_step's HOST logic on CPU (tests/test_chain_sim.py) and
the bench's data-parallel wiring under gloo (tests/test_fused_dp_cpu.py).
Layouts (transposed fragments, mask words) go through the same torch
oracles the GPU tests assert the kernels bit-match."""

import torch

from ray_shuffling_data_loader_amd.ops.shuffle_ops import (
    relu_mask_words,
    t_frag_swizzle,
    t_frag_unswizzle,
)


def _swz32(t, pi16=False):
    """t_frag_swizzle with rows padded to a 32-multiple — the kernels'
    slab granularity (the torch helper alone pads only to 16)."""
    m = t.shape[0]
    mp = (m + 31) // 32 * 32
    if mp != m:
        t = torch.nn.functional.pad(t, (0, 0, 0, mp - m))
    return t_frag_swizzle(t, pi16)


class FakeHip:
    @staticmethod
    def fwd_chain_bf16(x, W1, b1, W2, b2, W3, b3, w4, b4, target=None,
                       xt_out=None, pi16=False):
        if xt_out is not None:
            m = x.shape[0]
            mp = (m + 31) // 32 * 32
            xt_out.copy_(t_frag_swizzle(
                torch.nn.functional.pad(x, (0, 28, 0, mp - m)), pi16))
        if W1.shape[1] == 112:
            W1 = W1[:, :100]
        a1 = torch.relu(x.float() @ W1.float().t() + b1.float())
        a2 = torch.relu(a1 @ W2.float().t() + b2.float())
        a3 = torch.relu(a2 @ W3.float().t() + b3.float())
        out = a3 @ w4.float().unsqueeze(1) + b4.float()
        a1b, a2b = a1.bfloat16(), a2.bfloat16()
        res = (_swz32(a1b, pi16), relu_mask_words(a1b.float()),
               _swz32(a2b, pi16), relu_mask_words(a2b.float()),
               a3.bfloat16(), out.bfloat16())
        if target is None:
            return res
        m = x.shape[0]
        diff = res[5].float() - target.float().reshape(-1, 1)
        dyb = ((2.0 / m) * diff).bfloat16()
        return res + (dyb, diff.square().sum().reshape(1))

    @staticmethod
    def bwd_chain_bf16(dy, a3, mask1, mask2, w4, W3, W2, pi16=False):
        m_rows = dy.shape[0]

        def mask_of(words, n):
            mt = words.shape[0]
            w = words.to(torch.int64) & 0xFFFFFFFF
            sh = torch.arange(32, device=words.device)
            return (
                ((w.view(mt, 1, n) >> sh.view(1, 32, 1)) & 1)
                .reshape(mt * 32, n)[:m_rows]
                .bool()
            )

        da3 = dy.float() @ w4.float().unsqueeze(0)
        dz3 = (da3 * (a3.float() > 0)).bfloat16()
        da2 = dz3.float() @ W3.float()
        dz2 = (da2 * mask_of(mask2, 256)).bfloat16()
        da1 = dz2.float() @ W2.float()
        dz1 = (da1 * mask_of(mask1, 512)).bfloat16()
        dw4 = dy.float().t() @ a3.float()
        return (_swz32(dz1, pi16), _swz32(dz2, pi16),
                _swz32(dz3, pi16),
                dz1.float().sum(0), dz2.float().sum(0),
                dz3.float().sum(0), dy.float().sum(0), dw4)

    @staticmethod
    def swizzle_xt_bf16(x, pi16=False):
        m = x.shape[0]
        mp = (m + 31) // 32 * 32
        return t_frag_swizzle(
            torch.nn.functional.pad(x, (0, 28, 0, mp - m)), pi16)

    @staticmethod
    def wgrad_frag_bf16(at_f, bt_f, n, k, mchunks, nt_w, kt_w):
        m = mchunks * 16
        dz = t_frag_unswizzle(at_f, m, n)
        src = t_frag_unswizzle(bt_f, m, k)
        return dz.t().float() @ src.float()
