"""Trainer + Assigner integration on CPU/gloo, world_size=2: full AdaQP
adaptive pipeline (cost-model profiling, variance tracing, HiGHS MILP,
buffer rebuild) on a tiny synthetic reddit."""
import os
import types

import pytest
import torch
import torch.multiprocessing as mp

P = 2


def _args(tmp, mode, scheme, epochs, dataset='reddit', model='gcn'):
    return types.SimpleNamespace(
        dataset=dataset, model_name=model, mode=mode, assign_scheme=scheme,
        backend='gloo', init_method='env://', logger_level='WARNING',
        partition_dir=os.path.join(tmp, 'parts'), num_epochs=epochs,
        lr=0.01, log_steps=100, seed=1, scale=0.001)


def _worker(rank, world, port, tmp, mode, scheme, q, dataset='reddit',
            model='gcn'):
    os.environ.update(MASTER_ADDR='127.0.0.1', MASTER_PORT=str(port),
                      RANK=str(rank), WORLD_SIZE=str(world))
    from adaqp_amd.comm import Communicator
    from adaqp_amd.runtime.trainer import Trainer
    from adaqp_amd.helpers import AssignScheme
    try:
        t = Trainer(_args(tmp, mode, scheme, epochs=7, dataset=dataset,
                          model=model))
        t.assign_cycle = 3   # force a mid-run reassignment
        best = t.train()
        t.save(root=os.path.join(tmp, 'exp'))
        # adaptive runs must produce a non-degenerate mixed assignment store
        n_plans = len(t.engine.plans)
        q.put((rank, best['test'], n_plans))
    finally:
        Communicator.shutdown()


CASES = [
    ('Vanilla', None, 'reddit', 'gcn'),
    ('AdaQP', 'adaptive', 'reddit', 'gcn'),
    ('AdaQP-q', 'random', 'reddit', 'gcn'),
    ('AdaQP-p', None, 'reddit', 'gcn'),
    ('AdaQP', 'adaptive', 'yelp', 'sage'),   # multilabel micro-F1 distributed
]


@pytest.mark.parametrize('mode,scheme,dataset,model', CASES)
def test_trainer_end_to_end(tmp_path, mode, scheme, dataset, model):
    ctx = mp.get_context('spawn')
    q = ctx.SimpleQueue()
    port = 29490 + CASES.index((mode, scheme, dataset, model))
    procs = [ctx.Process(target=_worker,
                         args=(r, P, port, str(tmp_path), mode, scheme, q,
                               dataset, model))
             for r in range(P)]
    for p in procs:
        p.start()
    for p in procs:
        p.join(300)
        if p.is_alive():
            p.terminate()
            p.join(10)
            raise AssertionError('trainer worker hung')
        assert p.exitcode == 0
    results = []
    while not q.empty():
        results.append(q.get())
    assert len(results) == P
    for rank, test_acc, n_plans in results:
        assert 0.0 <= test_acc <= 1.0
        if mode in ('AdaQP', 'AdaQP-q'):
            assert n_plans == 5   # forward0..2 + backward1..2
    # artifacts written
    exp = os.path.join(str(tmp_path), 'exp', dataset, f'{P}part', model)
    assert os.path.isdir(os.path.join(exp, 'metrics'))
