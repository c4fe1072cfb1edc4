"""Unit tests for schema/pack/unpack/gather/partition ops (CPU reference
implementations; the GPU kernels are tested against these in
test_gpu_kernels.py)."""

import torch

from ray_shuffling_data_loader_amd.ops.shuffle_ops import (
    gather_rows,
    pack_columns,
    partition_rows,
    unpack_permute,
)
from ray_shuffling_data_loader_amd.utils.schema import (
    ColumnSpec,
    Schema,
    homogeneous_dtype,
)


def make_schema_het():
    return Schema(
        [
            ColumnSpec("a", torch.int64, 1),
            ColumnSpec("b", torch.float32, 1),
            ColumnSpec("c", torch.float64, 1),
            ColumnSpec("d", torch.float32, 4),
        ]
    )


def test_schema_layout_alignment():
    s = make_schema_het()
    # Descending dtype size: int64/float64 first, then float32s.
    for spec in s.columns:
        off = s.offsets[spec.name]
        assert off % min(8, spec.row_bytes // spec.numel) == 0
    assert s.row_stride % 16 == 0
    assert s.payload_bytes == 8 + 8 + 4 + 16
    assert s.row_stride == 48


def test_schema_homogeneous():
    s = Schema([ColumnSpec("x", torch.float32, 100),
                ColumnSpec("y", torch.float32, 1)])
    assert homogeneous_dtype(s) == torch.float32
    assert s.offsets["x"] == 0
    assert s.offsets["y"] == 400
    assert s.row_stride == 416
    assert homogeneous_dtype(make_schema_het()) is None


def rand_columns(n, schema, seed=0):
    g = torch.Generator().manual_seed(seed)
    cols = {}
    for spec in schema.columns:
        shape = (n,) if spec.numel == 1 else (n, spec.numel)
        if spec.dtype.is_floating_point:
            cols[spec.name] = torch.randn(shape, generator=g).to(spec.dtype)
        else:
            cols[spec.name] = torch.randint(
                0, 1000, shape, generator=g
            ).to(spec.dtype)
    return cols


def test_pack_unpack_roundtrip():
    schema = make_schema_het()
    n = 1000
    cols = rand_columns(n, schema)
    packed = pack_columns(cols, schema)
    assert packed.shape == (n, schema.row_stride)
    out = unpack_permute(packed, schema)
    for name in cols:
        assert torch.equal(
            out[name].reshape(cols[name].shape), cols[name]
        ), name


def test_unpack_with_permutation():
    schema = make_schema_het()
    n = 500
    cols = rand_columns(n, schema)
    packed = pack_columns(cols, schema)
    perm = torch.randperm(n)
    out = unpack_permute(packed, schema, perm=perm)
    for name in cols:
        expected = cols[name][perm]
        assert torch.equal(out[name].reshape(expected.shape), expected), name


def test_unpack_with_cast():
    schema = Schema([ColumnSpec("x", torch.float64, 1)])
    n = 100
    cols = {"x": torch.randn(n, dtype=torch.float64)}
    packed = pack_columns(cols, schema)
    out = unpack_permute(
        packed, schema, out_dtypes={"x": torch.float32}
    )
    assert out["x"].dtype == torch.float32
    assert torch.allclose(out["x"], cols["x"].float())


def test_gather_rows_cpu():
    n, stride = 200, 32
    src = torch.randint(0, 256, (n, stride), dtype=torch.uint8)
    perm = torch.randperm(n)
    out = gather_rows(src, perm)
    assert torch.equal(out, src[perm])


def test_partition_rows_cpu():
    n, stride = 300, 16
    src = torch.randint(0, 256, (n, stride), dtype=torch.uint8)
    dest = torch.randint(0, 4, (n,))
    grouped, counts = partition_rows(src, dest, 4)
    assert counts.sum() == n
    # Rows grouped by destination, stable order within each group.
    off = 0
    for d in range(4):
        expected = src[dest == d]
        got = grouped[off : off + counts[d]]
        assert torch.equal(got, expected), d
        off += int(counts[d])


def test_pack_with_scatter_perm():
    # pack_columns with perm scatters input row i to packed row perm[i].
    schema = Schema([ColumnSpec("x", torch.float32, 1)])
    n = 64
    cols = {"x": torch.arange(n, dtype=torch.float32)}
    perm = torch.randperm(n)
    packed = pack_columns(cols, schema, perm=perm)
    out = unpack_permute(packed, schema)
    assert torch.equal(out["x"][perm], cols["x"])


def test_t_frag_swizzle_roundtrip_shapes():
    """Round-trip through the wgrad fragment layout at awkward shapes."""
    import torch

    from ray_shuffling_data_loader_amd.ops.shuffle_ops import (
        t_frag_swizzle,
        t_frag_unswizzle,
    )

    for m, c in [(16, 32), (48, 64), (100, 128), (333, 256), (1024, 512)]:
        t = torch.randn(m, c).bfloat16()
        flat = t_frag_swizzle(t)
        mp = (m + 15) // 16 * 16
        assert flat.numel() == (c // 32) * (mp // 16) * 512
        back = t_frag_unswizzle(flat, m, c)
        assert torch.equal(back, t), (m, c)
        # pad rows must be zero (wgrad contributions)
        if mp != m:
            full = t_frag_unswizzle(flat, mp, c)
            assert torch.all(full[m:] == 0), (m, c)


def test_relu_mask_words_bits():
    import torch

    from ray_shuffling_data_loader_amd.ops.shuffle_ops import (
        relu_mask_words,
    )

    for m, n in [(32, 32), (70, 64), (129, 96)]:
        a = torch.randn(m, n)
        w = relu_mask_words(a)
        mt = (m + 31) // 32
        assert w.shape == (mt, n) and w.dtype == torch.int32
        bits = (w.to(torch.int64) & 0xFFFFFFFF).view(mt, 1, n)
        sh = torch.arange(32).view(1, 32, 1)
        got = ((bits >> sh) & 1).reshape(mt * 32, n)[:m].bool()
        assert torch.equal(got, a > 0), (m, n)
        # pad bits zero
        rest = ((bits >> sh) & 1).reshape(mt * 32, n)[m:]
        assert torch.all(rest == 0)


def test_t_frag_swizzle_pi16_roundtrip_and_invariance():
    """pi16 layout: swizzle/unswizzle round-trips, and the wgrad
    contraction is invariant when BOTH operands share the permutation
    (M is the contraction dim)."""
    import torch

    from ray_shuffling_data_loader_amd.ops.shuffle_ops import (
        _PI16_IDX,
        t_frag_swizzle,
        t_frag_unswizzle,
    )

    # involution with bits 2<->3 swapped
    assert [_PI16_IDX[p] for p in _PI16_IDX] == list(range(16))
    x = torch.randn(50, 64)
    for pi in (False, True):
        r = t_frag_unswizzle(t_frag_swizzle(x, pi), 50, 64, pi)
        assert torch.equal(r, x), pi
    # mixing flags must scramble rows (the layouts really differ)
    r = t_frag_unswizzle(t_frag_swizzle(x, True), 50, 64, False)
    assert not torch.equal(r, x)

    a, b = torch.randn(50, 32).bfloat16(), torch.randn(50, 64).bfloat16()
    d = {}
    for pi in (False, True):
        sa, sb = t_frag_swizzle(a, pi), t_frag_swizzle(b, pi)
        d[pi] = (
            t_frag_unswizzle(sa, 64, 32).float().t()
            @ t_frag_unswizzle(sb, 64, 64).float()
        )
    assert torch.allclose(d[False], d[True], atol=1e-4)
