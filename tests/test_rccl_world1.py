"""RCCL-backend execution at world_size=1 (the most that runs on a
1-GPU lease: NCCL/RCCL forbids two ranks on one device). Exercises the
exact collectives the 8-GPU run issues — all_to_all_single (payload +
bf16 params shapes), flat-bucket all_reduce, broadcast — through the
real RCCL library on device tensors, via the same Communicator methods
the training step calls."""
import os

import pytest
import torch

pytestmark = pytest.mark.gpu


def _comm():
    os.environ.update(MASTER_ADDR='127.0.0.1', MASTER_PORT='29581',
                      RANK='0', WORLD_SIZE='1', LOCAL_RANK='0')
    from adaqp_amd.comm import Communicator
    return Communicator()   # cpu:gloo,cuda:nccl on a GPU box


def test_rccl_collectives_world1():
    from adaqp_amd.comm import Communicator
    comm = _comm()
    try:
        assert 'nccl' in comm.backend
        dev = comm.device
        # all_to_all_v self-exchange: fp32 rows and the quantized wire
        # shapes (uint8 payload + bf16 params)
        x = torch.randn(37, 8, device=dev)
        out, _ = comm.exchange_rows(x, [37], [37])
        torch.cuda.synchronize()
        assert torch.equal(out, x)
        pay = torch.randint(0, 255, (4096,), dtype=torch.uint8, device=dev)
        pout = torch.empty_like(pay)
        comm.all_to_all_v(pout, pay, [4096], [4096])
        par = torch.randn(74, device=dev).to(torch.bfloat16)
        parout = torch.empty_like(par)
        comm.all_to_all_v(parout, par, [74], [74])
        torch.cuda.synchronize()
        assert torch.equal(pout, pay) and torch.equal(parout, par)
        # flat grad all-reduce + broadcast through RCCL
        lin = torch.nn.Linear(8, 8).to(dev)
        lin(torch.randn(4, 8, device=dev)).sum().backward()
        g0 = lin.weight.grad.clone()
        comm.flat_all_reduce_grads(lin.parameters())
        torch.cuda.synchronize()
        assert torch.allclose(lin.weight.grad, g0)
        comm.sync_model_params(lin)
        # reductions used by metrics/timing
        t = torch.tensor([3.0], device=dev)
        comm.all_reduce_sum(t)
        comm.all_reduce_max(t)
        comm.barrier()
        torch.cuda.synchronize()
        assert float(t.item()) == 3.0
    finally:
        Communicator.shutdown()
