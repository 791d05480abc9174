"""Property-based fuzz of the quantization oracle (hypothesis): for ANY
shape/bits/seed/value-range, pack->unpack must round-trip within the
per-bit error bound and payload sizes must match the wire contract."""
import numpy as np
import torch
from hypothesis import given, settings, strategies as st

from adaqp_amd.ops.quant import pack_torch, unpack_torch, bytes_per_node


@settings(max_examples=60, deadline=None)
@given(
    n=st.integers(1, 65),
    f=st.integers(1, 70),
    bits=st.sampled_from([2, 4, 8]),
    seed=st.integers(0, 2**31 - 1),
    scale=st.floats(1e-3, 1e3),
    offset=st.floats(-1e3, 1e3),
)
def test_roundtrip_bound_fuzz(n, f, bits, seed, scale, offset):
    gen = torch.Generator().manual_seed(seed & 0xFFFF)
    x = torch.rand(n, f, generator=gen) * scale + offset
    tag = torch.arange(n, dtype=torch.int64)
    payload, sc, rmin = pack_torch(x, bits, seed, node_tag=tag)
    assert payload.shape == (n, bytes_per_node(f, bits))
    y = unpack_torch(payload, bits, sc, rmin, f)
    # max error: one quantization step (stochastic rounding) + bf16
    # rounding of scale/rmin
    rng = (x.max(dim=1).values - x.min(dim=1).values).clamp(min=0)
    step = rng / (2.0 ** bits - 1)
    bf16_slop = (rng + x.abs().max(dim=1).values) * 2 ** -8
    bound = (step + bf16_slop + 1e-6)[:, None]
    assert bool(((y - x).abs() <= bound).all()), \
        float(((y - x).abs() - bound).max())


@settings(max_examples=30, deadline=None)
@given(
    n=st.integers(1, 40),
    f=st.integers(1, 40),
    bits=st.sampled_from([2, 4, 8]),
)
def test_constant_rows_exact_fuzz(n, f, bits):
    """Constant rows (range 0) must reproduce exactly via rmin."""
    x = torch.full((n, f), 3.25)   # exactly representable in bf16
    tag = torch.arange(n, dtype=torch.int64)
    payload, sc, rmin = pack_torch(x, bits, 7, node_tag=tag)
    y = unpack_torch(payload, bits, sc, rmin, f)
    assert torch.equal(y, x)
