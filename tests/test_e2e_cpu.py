"""End-to-end distributed correctness on CPU/gloo, world_size=2
(BASELINE.json config #1: plumbing without a GPU).

For fp32 modes the partitioned forward/backward must match a dense
global-graph reference EXACTLY (SURVEY.md §7 'hard parts': transposed
backward normalizations). Quantized modes are checked for bounded error.
"""
import os

import pytest
import torch
import torch.distributed as dist
import torch.multiprocessing as mp

from adaqp_amd.graph import tiny_ring_graph, random_partitioned_graph, partition_all
from adaqp_amd.helpers import DistGNNType, RunMode

P = 2


def dense_reference(g, model_kind, agg_type, weights, num_layers, hidden):
    """Dense global-graph forward with the same math (autograd-enabled)."""
    N = g.num_nodes
    A = torch.zeros(N, N)
    A[g.dst, g.src] = 1.0
    din = torch.bincount(g.dst, minlength=N).float().clamp(min=1)
    dout = torch.bincount(g.src, minlength=N).float().clamp(min=1)
    h = g.feats
    for i in range(num_layers):
        if model_kind == DistGNNType.DistGCN:
            agg = (A @ (h * dout.pow(-0.5)[:, None])) * din.pow(-0.5)[:, None]
            h = agg @ weights[f'convs.{i}.linear.weight'] + weights[f'convs.{i}.linear.bias']
        else:
            if agg_type == 'mean':
                agg = (A @ h) / din[:, None]
                h = h @ weights[f'convs.{i}.fc_self.weight'] + weights[f'convs.{i}.fc_self.bias'] \
                    + agg @ weights[f'convs.{i}.fc_neigh.weight'] + weights[f'convs.{i}.fc_neigh.bias']
            else:
                agg = (A @ h + h) / (din + 1)[:, None]
                h = agg @ weights[f'convs.{i}.fc_neigh.weight'] + weights[f'convs.{i}.fc_neigh.bias']
        if i < num_layers - 1:
            h = torch.nn.functional.layer_norm(
                h, (hidden,), weights[f'norms.{i}.weight'], weights[f'norms.{i}.bias'])
            h = torch.relu(h)
    return h


def _build_model(model_kind, agg_type, in_dim, hidden, out_dim, L):
    from adaqp_amd.models import DistGCN, DistSAGE
    torch.manual_seed(7)
    if model_kind == DistGNNType.DistGCN:
        return DistGCN(in_dim, hidden, out_dim, num_layers=L, dropout=0.0)
    return DistSAGE(in_dim, hidden, out_dim, num_layers=L, dropout=0.0,
                    aggregator_type=agg_type)


def _worker(rank, world, port, model_kind, agg_type, mode, q):
    os.environ['MASTER_ADDR'] = '127.0.0.1'
    os.environ['MASTER_PORT'] = str(port)
    os.environ['RANK'] = str(rank)
    os.environ['WORLD_SIZE'] = str(world)
    from adaqp_amd.comm import Communicator
    from adaqp_amd.runtime import GraphEngine
    comm = Communicator(backend='gloo')
    try:
        torch.manual_seed(7)
        g = tiny_ring_graph(40, feat_dim=6, num_classes=3, extra_edges=60)
        parts = partition_all(g, world)
        lg = parts[rank]
        in_dim, hidden, out_dim, L = 6, 8, 3, 3
        engine = GraphEngine(lg, RunMode(mode), model_kind,
                             msg_dims=[in_dim] + [hidden] * (L - 1),
                             agg_type=agg_type, device=torch.device('cpu'))
        if engine.bit_type.name == 'QUANT':
            engine.set_uniform_assignment(8)
        model = _build_model(model_kind, agg_type, in_dim, hidden, out_dim, L)
        comm.sync_model_params(model)
        model.eval()  # kill dropout; exchange still honors mode via is_train flag
        model.train()
        model.dropout.p = 0.0

        logits = model(engine, lg.feats)
        # global-mean loss over ALL nodes so grads are comparable
        n_total = torch.tensor([float(lg.num_inner)])
        comm.all_reduce_sum(n_total)
        loss = torch.nn.functional.cross_entropy(
            logits, lg.labels, reduction='sum') / n_total
        loss.backward()
        comm.flat_all_reduce_grads(model.parameters())

        weights = {k: v.detach().clone() for k, v in model.named_parameters()}
        ref_h = dense_reference(g, model_kind, agg_type, weights, L, hidden)
        ref_logits_local = ref_h[lg.local_to_global[:lg.num_inner]]
        fwd_err = (logits - ref_logits_local).abs().max().item()

        # dense grad reference
        for k, v in weights.items():
            v.requires_grad_(True)
        ref_h2 = dense_reference(g, model_kind, agg_type, weights, L, hidden)
        ref_loss = torch.nn.functional.cross_entropy(
            ref_h2, g.labels, reduction='sum') / n_total
        ref_loss.backward()
        grad_err = max((weights[k].grad - p.grad).abs().max().item()
                       for k, p in model.named_parameters())
        q.put((rank, fwd_err, grad_err))
    finally:
        Communicator.shutdown()


CASES = [
    (DistGNNType.DistGCN, 'mean', 'Vanilla'),
    (DistGNNType.DistGCN, 'mean', 'AdaQP-p'),
    (DistGNNType.DistSAGE, 'mean', 'Vanilla'),
    (DistGNNType.DistSAGE, 'gcn', 'Vanilla'),
    (DistGNNType.DistSAGE, 'mean', 'AdaQP-p'),
]


def _run_pair(target, args_fn, port):
    ctx = mp.get_context('spawn')
    q = ctx.SimpleQueue()
    procs = [ctx.Process(target=target, args=args_fn(r, q)) for r in range(P)]
    for p in procs:
        p.start()
    results = []
    for p in procs:
        p.join(180)
        if p.is_alive():
            p.terminate()
            p.join(10)
            raise AssertionError('worker hung')
        assert p.exitcode == 0, f'worker exit {p.exitcode}'
    while not q.empty():
        results.append(q.get())
    assert len(results) == P
    return results


@pytest.mark.parametrize('model_kind,agg_type,mode', CASES)
def test_matches_dense_reference(model_kind, agg_type, mode):
    port = 29600 + CASES.index((model_kind, agg_type, mode))
    results = _run_pair(_worker,
                        lambda r, q: (r, P, port, model_kind, agg_type, mode, q),
                        port)
    for rank, fwd_err, grad_err in results:
        assert fwd_err < 1e-4, f'rank {rank} forward err {fwd_err}'
        assert grad_err < 1e-4, f'rank {rank} grad err {grad_err}'


def _quant_worker(rank, world, port, q):
    os.environ['MASTER_ADDR'] = '127.0.0.1'
    os.environ['MASTER_PORT'] = str(port)
    os.environ['RANK'] = str(rank)
    os.environ['WORLD_SIZE'] = str(world)
    from adaqp_amd.comm import Communicator
    from adaqp_amd.runtime import GraphEngine
    from adaqp_amd.ops.dist_agg import fp_exchange, qt_exchange
    comm = Communicator(backend='gloo')
    try:
        torch.manual_seed(3)
        g = random_partitioned_graph(120, 700, 16, 4, world, seed=5, cut_frac=0.4)
        lg = partition_all(g, world)[rank]
        engine = GraphEngine(lg, RunMode('AdaQP-q'), DistGNNType.DistGCN,
                             msg_dims=[16, 8, 8], device=torch.device('cpu'))
        engine.set_uniform_assignment(8)
        x = torch.randn(lg.num_inner, 16)
        fp = fp_exchange(engine, x, 'forward0')
        qt = qt_exchange(engine, x, 'forward0')
        err = (fp - qt).abs().max().item() if fp.numel() else 0.0
        # 8-bit stochastic quantization on randn: step ~ range/255
        q.put((rank, err))
    finally:
        Communicator.shutdown()


def test_quantized_exchange_error_bounded():
    results = _run_pair(_quant_worker, lambda r, q: (r, P, 29477, q), 29477)
    for rank, err in results:
        assert err < 0.12, f'rank {rank} quant exchange err {err}'
