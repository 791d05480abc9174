"""Edge cases: peer pairs with zero boundary, assigner score math."""
import os

import torch
import torch.multiprocessing as mp

from adaqp_amd.graph import GlobalGraph, build_local_graph, global_degrees
from adaqp_amd.graph.synthetic import _dedup_edges


def _line_graph(parts=3, per=10, feat=4):
    """Parts arranged in a line: parts 0 and 2 share no edges."""
    n = parts * per
    s, d = [], []
    for i in range(n - 1):
        if abs(i // per - (i + 1) // per) <= 1:
            s += [i, i + 1]
            d += [i + 1, i]
    i = torch.arange(n)
    s = torch.cat([torch.tensor(s), i])
    d = torch.cat([torch.tensor(d), i])
    s, d = _dedup_edges(s, d, n)
    g = torch.Generator().manual_seed(0)
    return GlobalGraph(n, s, d, torch.randn(n, feat, generator=g),
                       torch.randint(0, 3, (n,), generator=g),
                       torch.ones(n, dtype=torch.bool),
                       torch.zeros(n, dtype=torch.bool),
                       torch.zeros(n, dtype=torch.bool), 3, False)


def test_disconnected_pair_structure():
    g = _line_graph()
    assign = torch.arange(30) // 10
    for r in range(3):
        lg = build_local_graph(g, assign, r, 3)
        lg.validate()
        if r == 0:
            assert lg.send_splits[2] == 0 and lg.recv_splits[2] == 0
        if r == 2:
            assert lg.send_splits[0] == 0 and lg.recv_splits[0] == 0


def _worker(rank, world, port, q):
    os.environ.update(MASTER_ADDR='127.0.0.1', MASTER_PORT=str(port),
                      RANK=str(rank), WORLD_SIZE=str(world))
    from adaqp_amd.comm import Communicator
    from adaqp_amd.runtime import GraphEngine
    from adaqp_amd.ops.dist_agg import fp_exchange, qt_exchange
    from adaqp_amd.helpers import RunMode, DistGNNType
    comm = Communicator(backend='gloo')
    try:
        g = _line_graph()
        assign = torch.arange(30) // 10
        lg = build_local_graph(g, assign, rank, world)
        engine = GraphEngine(lg, RunMode('AdaQP-q'), DistGNNType.DistGCN,
                             msg_dims=[4, 4, 4], device=torch.device('cpu'))
        engine.set_uniform_assignment(4)
        x = torch.randn(lg.num_inner, 4)
        fp = fp_exchange(engine, x, 'forward0')
        qt = qt_exchange(engine, x, 'forward0')
        err = (fp - qt).abs().max().item() if fp.numel() else 0.0
        q.put((rank, err))
    finally:
        Communicator.shutdown()


def test_disconnected_pair_exchange():
    """Exchange with a zero-volume channel (ranks 0<->2) must not hang
    and must keep the other channels correct."""
    ctx = mp.get_context('spawn')
    q = ctx.SimpleQueue()
    procs = [ctx.Process(target=_worker, args=(r, 3, 29466, q))
             for r in range(3)]
    for p in procs:
        p.start()
    for p in procs:
        p.join(120)
        if p.is_alive():
            p.terminate()
            raise AssertionError('hung on zero-volume channel')
        assert p.exitcode == 0
    n = 0
    while not q.empty():
        _, err = q.get()
        assert err < 0.5
        n += 1
    assert n == 3


def test_edge_score_matches_bruteforce():
    from adaqp_amd.assigner.assigner import Assigner
    from adaqp_amd.runtime.engine import GraphEngine
    from adaqp_amd.helpers import RunMode, DistGNNType
    from adaqp_amd.graph import random_partitioned_graph
    g = random_partitioned_graph(60, 400, 4, 3, 1, seed=3)
    assign = torch.zeros(60, dtype=torch.int64)
    lg = build_local_graph(g, assign, 0, 1)
    engine = GraphEngine.__new__(GraphEngine)
    engine.graph = lg
    a = Assigner.__new__(Assigner)
    a.engine = engine
    src = torch.rand(lg.num_nodes) + 0.5
    dst = torch.rand(lg.num_inner) + 0.5
    score = a._edge_score(src, dst)
    # brute force
    ref = torch.zeros(lg.num_nodes)
    for r in range(lg.num_inner):
        for e in range(int(lg.indptr[r]), int(lg.indptr[r + 1])):
            ref[lg.indices[e]] += dst[r]
    ref = ref * src
    assert torch.allclose(score, ref, atol=1e-5)
