"""8-rank rehearsal of the flagship bench.

The driver's headline run is ``bench.py --gpus 8`` on an 8-GPU node; the
first time that executes must not be the first time the 8-rank path runs.
These tests exercise the COMPLETE 8-rank pipeline on CPU/gloo — per-rank
partition build, quantized mixed-bit exchange with 7 peers, the adaptive
profiler + MILP, grad all-reduce, metric aggregation — via the exact
torch.distributed.run incantation the driver uses.

Plus the nccl-only edge case we cannot execute without 8 GPUs, reduced to
its tensor-shape essence on gloo: a world where EVERY peer split is zero
(fully isolated partitions -> 0-numel all_to_all payloads).
"""
import json
import os
import subprocess
import sys

import torch
import torch.multiprocessing as mp

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def _run_bench_8(tmp_path, extra, port):
    cmd = [sys.executable, '-m', 'torch.distributed.run', '--nnodes=1',
           '--nproc-per-node=8', '--master-addr', '127.0.0.1',
           '--master-port', str(port), 'bench.py', '--cpu',
           '--gpus', '8', '--steps', '2', '--warmup', '1',
           '--scale', '0.004', '--part-dir', str(tmp_path / 'parts')] + extra
    out = subprocess.run(cmd, cwd=REPO, capture_output=True, text=True,
                         timeout=900)
    assert out.returncode == 0, out.stderr[-3000:]
    lines = [l for l in out.stdout.splitlines() if l.startswith('{')]
    assert len(lines) == 1, out.stdout
    return json.loads(lines[0])


def test_bench_8rank_adaqp(tmp_path):
    """The headline config shape: 8 partitions, AdaQP mode, quantized
    exchange active in the timed region on every rank."""
    d = _run_bench_8(tmp_path, [], 29541)
    assert d['n_gpus'] == 8
    assert '8-part' in d['metric']
    assert 'ogbn-products' in d['metric']
    assert d['config']['parallelism'] == 'graph-partition dp8'
    assert d['value'] > 0


def test_bench_8rank_adaptive(tmp_path):
    """Adaptive scheme at 8 ranks: cost profiler (p2p send/recv with 7
    peers) + HiGHS MILP + scatter, end-to-end."""
    d = _run_bench_8(tmp_path, ['--assign-scheme', 'adaptive'], 29542)
    assert d['n_gpus'] == 8
    assert d['config']['mode'] == 'AdaQP'


def _isolated_worker(rank, world, port, q):
    os.environ.update(MASTER_ADDR='127.0.0.1', MASTER_PORT=str(port),
                      RANK=str(rank), WORLD_SIZE=str(world))
    from adaqp_amd.comm import Communicator
    from adaqp_amd.runtime import GraphEngine
    from adaqp_amd.ops.dist_agg import fp_exchange, qt_exchange
    from adaqp_amd.helpers import RunMode, DistGNNType
    from adaqp_amd.graph import GlobalGraph, build_local_graph
    from adaqp_amd.graph.synthetic import _dedup_edges
    comm = Communicator(backend='gloo')
    try:
        # two fully disconnected cliques -> zero boundary nodes anywhere
        n, per = 12, 6
        s, d = [], []
        for blk in range(2):
            for i in range(per):
                for j in range(per):
                    s.append(blk * per + i)
                    d.append(blk * per + j)
        s, d = _dedup_edges(torch.tensor(s), torch.tensor(d), n)
        gen = torch.Generator().manual_seed(0)
        g = GlobalGraph(n, s, d, torch.randn(n, 4, generator=gen),
                        torch.randint(0, 3, (n,), generator=gen),
                        torch.ones(n, dtype=torch.bool),
                        torch.zeros(n, dtype=torch.bool),
                        torch.zeros(n, dtype=torch.bool), 3, False)
        assign = torch.arange(n) // per
        lg = build_local_graph(g, assign, rank, world)
        assert lg.num_remote == 0 and sum(lg.recv_splits) == 0
        engine = GraphEngine(lg, RunMode('AdaQP-q'), DistGNNType.DistGCN,
                             msg_dims=[4, 4], device=torch.device('cpu'))
        engine.set_uniform_assignment(4)
        x = torch.randn(lg.num_inner, 4)
        fp = fp_exchange(engine, x, 'forward0')    # 0-numel payloads
        qt = qt_exchange(engine, x, 'forward0')
        assert fp.shape[0] == 0 and qt.shape[0] == 0
        q.put(rank)
    finally:
        Communicator.shutdown()


def test_all_splits_zero_exchange():
    """Every split zero on every rank: the all_to_all payloads are
    0-numel tensors. Must complete (not hang, not throw) — this is the
    shape nccl sees when two partitions share no edges at world=8."""
    ctx = mp.get_context('spawn')
    q = ctx.SimpleQueue()
    procs = [ctx.Process(target=_isolated_worker, args=(r, 2, 29467, q))
             for r in range(2)]
    for p in procs:
        p.start()
    for p in procs:
        p.join(120)
        if p.is_alive():
            p.terminate()
            raise AssertionError('hung on all-zero-splits exchange')
        assert p.exitcode == 0
    done = 0
    while not q.empty():
        q.get()
        done += 1
    assert done == 2
