"""Multilabel (yelp/amazonProducts-style) loss + micro-F1 path, world=1."""
import os

import torch

from adaqp_amd.comm import Communicator
from adaqp_amd.runtime import GraphEngine
from adaqp_amd.runtime.utils import train_epoch, evaluate, global_train_count
from adaqp_amd.models import DistSAGE
from adaqp_amd.helpers import RunMode, DistGNNType
from adaqp_amd.graph import random_partitioned_graph, partition_all


def test_multilabel_training(monkeypatch):
    monkeypatch.setenv('MASTER_ADDR', '127.0.0.1')
    monkeypatch.setenv('MASTER_PORT', '29460')
    monkeypatch.setenv('RANK', '0')
    monkeypatch.setenv('WORLD_SIZE', '1')
    comm = Communicator(backend='gloo')
    try:
        g = random_partitioned_graph(300, 2500, 12, 6, 1, seed=8,
                                     multilabel=True)
        lg = partition_all(g, 1)[0]
        engine = GraphEngine(lg, RunMode('Vanilla'), DistGNNType.DistSAGE,
                             msg_dims=[12, 16, 16], device=torch.device('cpu'))
        model = DistSAGE(12, 16, 6, 3, dropout=0.0)
        opt = torch.optim.Adam(model.parameters(), lr=0.01)
        gc = global_train_count(engine)
        l0 = float(train_epoch(engine, model, opt, gc, multilabel=True))
        for _ in range(30):
            l = float(train_epoch(engine, model, opt, gc, multilabel=True))
        assert l < l0
        m = evaluate(engine, model, multilabel=True)
        for k in ('train', 'val', 'test'):
            assert 0.0 <= m[k] <= 1.0
    finally:
        Communicator.shutdown()


def test_multilabel_teacher_labels_learnable(monkeypatch):
    """Teacher-thresholded multilabel targets must be LEARNABLE (micro-F1
    well above the all-negative 0.0 that random targets produce) so
    Vanilla-vs-AdaQP accuracy comparisons are meaningful on the
    multilabel datasets too."""
    import os
    import torch
    os.environ.update(MASTER_ADDR='127.0.0.1', MASTER_PORT='29596',
                      RANK='0', WORLD_SIZE='1', LOCAL_RANK='0')
    from adaqp_amd.comm import Communicator
    from adaqp_amd.runtime import GraphEngine
    from adaqp_amd.runtime.utils import train_epoch, evaluate, global_train_count
    from adaqp_amd.models import DistSAGE
    from adaqp_amd.helpers import RunMode, DistGNNType
    from adaqp_amd.graph import random_partitioned_graph, partition_all
    comm = Communicator(backend='gloo')
    try:
        g = random_partitioned_graph(3000, 36000, 32, 20, 1, seed=5,
                                     multilabel=True, teacher_labels=True)
        assert 0.05 < float(g.labels.mean()) < 0.15   # yelp-like density
        lg = partition_all(g, 1)[0]
        engine = GraphEngine(lg, RunMode('Vanilla'), DistGNNType.DistSAGE,
                             [32, 64, 64], device=torch.device('cpu'))
        model = DistSAGE(32, 64, 20, 3, dropout=0.0)
        opt = torch.optim.Adam(model.parameters(), lr=0.01)
        gc = global_train_count(engine)
        for _ in range(90):
            train_epoch(engine, model, opt, gc, True)
        f1 = evaluate(engine, model, True)['val']
        assert f1 > 0.3, f'multilabel teacher targets not learnable: {f1}'
    finally:
        Communicator.shutdown()
