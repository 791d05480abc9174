"""Quantization tests (SURVEY.md §4 test plan item 1): round-trip error
bound, stochastic unbiasedness, mixed-bit wire layout consistency."""
import torch
import pytest

from adaqp_amd.ops import quant as Q
from adaqp_amd.ops.kernels import mixed_quantize, mixed_dequantize
from adaqp_amd.comm.buffers import _layout, BITS_SET, bytes_per_node


@pytest.mark.parametrize('bits', [2, 4, 8])
@pytest.mark.parametrize('F', [7, 64, 256])
def test_roundtrip_error_bound(bits, F):
    torch.manual_seed(0)
    x = torch.randn(50, F)
    tag = torch.arange(50)
    payload, scale, rmin = Q.pack_torch(x, bits, seed=123, node_tag=tag)
    assert payload.shape == (50, bytes_per_node(F, bits))
    xhat = Q.unpack_torch(payload, bits, scale, rmin, F)
    rng = x.max(1).values - x.min(1).values
    step = rng / (2 ** bits - 1)
    err = (xhat - x).abs().max(dim=1).values
    # stochastic rounding error < 1 quantization step (+ bf16 param slack)
    assert (err <= step * 1.05 + rng * 0.01).all()


def test_constant_row():
    x = torch.full((3, 16), 2.5)
    payload, scale, rmin = Q.pack_torch(x, 4, seed=1, node_tag=torch.arange(3))
    assert (scale.float() == 0).all()
    xhat = Q.unpack_torch(payload, 4, scale, rmin, 16)
    assert torch.allclose(xhat, x, atol=0.02)


@pytest.mark.parametrize('bits', [2, 4])
def test_unbiasedness(bits):
    torch.manual_seed(1)
    x = torch.randn(8, 32)
    tag = torch.arange(8)
    acc = torch.zeros_like(x)
    K = 400
    for s in range(K):
        payload, scale, rmin = Q.pack_torch(x, bits, seed=s * 7919 + 13, node_tag=tag)
        acc += Q.unpack_torch(payload, bits, scale, rmin, 32)
    mean = acc / K
    rng = (x.max(1).values - x.min(1).values)[:, None]
    step = rng / (2 ** bits - 1)
    # E[dequant(quant(x))] = x within sampling noise (~step/sqrt(K)) + bf16 slack
    tol = step * (3.0 / K ** 0.5) + rng * 0.01 + 1e-3
    assert ((mean - x).abs() <= tol).all(), (mean - x).abs().max()


def test_noise_deterministic_and_uniform():
    u1 = Q.uniform_noise(42, torch.arange(100), 64)
    u2 = Q.uniform_noise(42, torch.arange(100), 64)
    assert torch.equal(u1, u2)
    u3 = Q.uniform_noise(43, torch.arange(100), 64)
    assert not torch.equal(u1, u3)
    assert 0.45 < u1.mean() < 0.55
    assert u1.min() >= 0 and u1.max() < 1


def _mk_plan(bits_vecs, rows_vecs, F):
    return _layout(bits_vecs, rows_vecs, F)


def test_mixed_wire_roundtrip():
    """Send plan on x, 'wire transfer', recv plan reconstructs rows."""
    torch.manual_seed(2)
    F = 24
    x = torch.randn(40, F)
    # two peers with mixed bit widths
    bits_p0 = torch.tensor([2, 8, 4, 4, 2])
    rows_p0 = torch.tensor([3, 7, 11, 20, 35])
    bits_p1 = torch.tensor([8, 2, 8])
    rows_p1 = torch.tensor([1, 2, 39])
    send = _mk_plan([bits_p0, bits_p1], [rows_p0, rows_p1], F)
    # receiver stores peer-major remote rows 0..7
    recv = _mk_plan([bits_p0, bits_p1],
                    [torch.arange(5), torch.arange(5, 8)], F)
    payload, params = mixed_quantize(x, send, seed=99)
    assert payload.numel() == send.total_bytes == recv.total_bytes
    out = torch.zeros(8, F)
    mixed_dequantize(payload, params, recv, out)
    ref_rows = torch.cat([rows_p0, rows_p1])
    bits_all = torch.cat([bits_p0, bits_p1])
    rng = (x[ref_rows].max(1).values - x[ref_rows].min(1).values)
    step = rng / (2.0 ** bits_all - 1)
    err = (out - x[ref_rows]).abs().max(1).values
    assert (err <= step * 1.05 + rng * 0.01).all()


def test_layout_rejects_bad_bits():
    with pytest.raises(ValueError):
        _layout([torch.tensor([3])], [torch.tensor([0])], 8)
