"""Fuzz the mixed-bit wire bookkeeping: repeated RANDOM per-node bit
assignments (the reassignment path, SURVEY.md §7 'fiddly' item); each
round the quantized exchange must reconstruct every node within its own
bit width's error bound. 2 ranks, CPU/gloo."""
import os

import torch
import torch.multiprocessing as mp

P = 2


def _worker(rank, world, port, q):
    os.environ.update(MASTER_ADDR='127.0.0.1', MASTER_PORT=str(port),
                      RANK=str(rank), WORLD_SIZE=str(world))
    from adaqp_amd.comm import Communicator
    from adaqp_amd.comm.buffers import BITS_SET
    from adaqp_amd.runtime import GraphEngine
    from adaqp_amd.ops.dist_agg import fp_exchange, qt_exchange
    from adaqp_amd.helpers import RunMode, DistGNNType
    from adaqp_amd.graph import random_partitioned_graph, partition_all
    comm = Communicator(backend='gloo')
    try:
        torch.manual_seed(100 + rank)
        g = random_partitioned_graph(300, 2500, 24, 4, world, seed=44,
                                     cut_frac=0.5)
        lg = partition_all(g, world)[rank]
        engine = GraphEngine(lg, RunMode('AdaQP-q'), DistGNNType.DistGCN,
                             msg_dims=[24, 24, 24],
                             device=torch.device('cpu'))
        gen = torch.Generator().manual_seed(7)   # same on both ranks? no:
        gen = torch.Generator().manual_seed(7 + rank)  # sender-side bits
        worst = 0.0
        for trial in range(5):
            assignments = {}
            for k in engine.exchange_keys():
                assignments[k] = {
                    p: torch.tensor(BITS_SET)[torch.randint(
                        0, len(BITS_SET), (n,), generator=gen)]
                    for p, n in enumerate(lg.send_splits) if n}
            engine.set_assignment(assignments)
            x = torch.randn(lg.num_inner, 24)
            for k in ('forward0', 'backward1'):
                fp = fp_exchange(engine, x, k)
                qt = qt_exchange(engine, x, k)
                if fp.numel() == 0:
                    continue
                # per received node: bound = range/(2^bits-1) + bf16 slack
                plan = engine.plans[k].recv
                for b in (2, 4, 8):
                    rows = plan.rows[b]
                    if rows.numel() == 0:
                        continue
                    ref = fp[rows]
                    got = qt[rows]
                    rng = ref.max(1).values - ref.min(1).values
                    step = rng / (2 ** b - 1)
                    err = (got - ref).abs().max(1).values
                    bad = (err > step * 1.05 + rng * 0.02 + 1e-4)
                    assert not bad.any(), (trial, k, b, float(err.max()))
                    worst = max(worst, float(err.max()))
        q.put((rank, worst))
    finally:
        Communicator.shutdown()


def test_random_reassignment_fuzz():
    ctx = mp.get_context('spawn')
    q = ctx.SimpleQueue()
    procs = [ctx.Process(target=_worker, args=(r, P, 29468, q))
             for r in range(P)]
    for p in procs:
        p.start()
    for p in procs:
        p.join(300)
        if p.is_alive():
            p.terminate()
            raise AssertionError('hung')
        assert p.exitcode == 0
    n = 0
    while not q.empty():
        q.get()
        n += 1
    assert n == P
