"""Convergence-parity check (the reference's correctness oracle,
SURVEY.md §4): on a learnable synthetic graph, AdaQP-q (mixed-bit
stochastic quantization) must reach test accuracy close to Vanilla.
2 ranks, CPU/gloo."""
import os

import pytest
import torch
import torch.multiprocessing as mp

P = 2


def _train(rank, world, port, mode, bits, q):
    os.environ.update(MASTER_ADDR='127.0.0.1', MASTER_PORT=str(port),
                      RANK=str(rank), WORLD_SIZE=str(world))
    from adaqp_amd.comm import Communicator
    from adaqp_amd.runtime import GraphEngine
    from adaqp_amd.runtime.utils import train_epoch, evaluate, global_train_count
    from adaqp_amd.models import DistGCN
    from adaqp_amd.helpers import RunMode, DistGNNType
    from adaqp_amd.graph import random_partitioned_graph, partition_all
    comm = Communicator(backend='gloo')
    try:
        torch.manual_seed(11)
        g = random_partitioned_graph(1500, 15000, 16, 5, world, seed=21,
                                     cut_frac=0.3, teacher_labels=True)
        lg = partition_all(g, world)[rank]
        engine = GraphEngine(lg, RunMode(mode), DistGNNType.DistGCN,
                             msg_dims=[16, 32, 32], device=torch.device('cpu'))
        if engine.bit_type.name == 'QUANT':
            engine.set_uniform_assignment(bits)
        torch.manual_seed(33)
        model = DistGCN(16, 32, 5, num_layers=3, dropout=0.0)
        comm.sync_model_params(model)
        opt = torch.optim.Adam(model.parameters(), lr=0.01)
        gc = global_train_count(engine)
        for _ in range(150):
            train_epoch(engine, model, opt, gc, False)
        acc = evaluate(engine, model, False)
        q.put((rank, acc['test']))
    finally:
        Communicator.shutdown()


def _run(mode, bits, port):
    ctx = mp.get_context('spawn')
    q = ctx.SimpleQueue()
    procs = [ctx.Process(target=_train, args=(r, P, port, mode, bits, q))
             for r in range(P)]
    for p in procs:
        p.start()
    for p in procs:
        p.join(600)
        if p.is_alive():
            p.terminate()
            raise AssertionError('hung')
        assert p.exitcode == 0
    accs = []
    while not q.empty():
        accs.append(q.get()[1])
    assert len(accs) == P and abs(accs[0] - accs[1]) < 1e-6
    return accs[0]


def test_quantized_convergence_parity():
    vanilla = _run('Vanilla', 0, 29430)
    quant8 = _run('AdaQP-q', 8, 29431)
    quant4 = _run('AdaQP-q', 4, 29432)
    assert vanilla > 0.5, f'teacher-labeled graph should be learnable, got {vanilla}'
    assert quant8 >= vanilla - 0.03, (vanilla, quant8)
    assert quant4 >= vanilla - 0.06, (vanilla, quant4)
