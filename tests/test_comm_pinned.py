"""Unit tests for the pinned staging cache and the spmm out= contract."""
import torch

from adaqp_amd.comm.communicator import Communicator


def _bare_comm():
    c = Communicator.__new__(Communicator)
    return c


def test_pinned_cache_grows_and_reuses():
    c = _bare_comm()
    a = c._pinned('send', (10, 4), torch.float32)
    assert a.shape == (10, 4)
    b = c._pinned('send', (5, 4), torch.float32)      # smaller: reuse
    assert b.data_ptr() == a.data_ptr()
    d = c._pinned('send', (100, 4), torch.float32)    # bigger: regrow
    assert d.numel() == 400
    e = c._pinned('recv', (10, 4), torch.float32)     # distinct tag
    assert e.data_ptr() != c._pinned('send', (10, 4), torch.float32).data_ptr()
    f = c._pinned('send', (10, 4), torch.uint8)       # distinct dtype
    assert f.dtype == torch.uint8


def test_spmm_out_param_cpu():
    from adaqp_amd.ops.kernels import SpmmView, spmm
    indptr = torch.tensor([0, 2, 3], dtype=torch.int64)
    indices = torch.tensor([0, 1, 2], dtype=torch.int64)
    v = SpmmView(indptr, indices, 0, 2)
    x = torch.randn(3, 4)
    out = torch.empty(2, 4)
    y = spmm(v, x, None, None, None, out=out)
    assert y.data_ptr() == out.data_ptr()
    ref = spmm(v, x, None, None, None)
    assert torch.allclose(y, ref)
