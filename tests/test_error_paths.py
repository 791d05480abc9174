"""Error-path behavior: loud, early, and specific failures (the
framework refuses to limp along silently)."""
import os
import subprocess
import sys

import numpy as np
import pytest
import torch

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def test_ingest_missing_adj(tmp_path):
    from adaqp_amd.graph import load_graph_dir
    with pytest.raises(FileNotFoundError, match='adj_full.npz'):
        load_graph_dir(str(tmp_path))


def test_ingest_shape_mismatch(tmp_path):
    sp = pytest.importorskip('scipy.sparse')
    m = sp.csr_matrix(np.eye(8, dtype=np.float32))
    sp.save_npz(os.path.join(tmp_path, 'adj_full.npz'), m)
    np.save(os.path.join(tmp_path, 'feats.npy'),
            np.zeros((5, 3), dtype=np.float32))   # 5 != 8 nodes
    from adaqp_amd.graph import load_graph_dir
    with pytest.raises(ValueError, match='feats rows'):
        load_graph_dir(str(tmp_path))


def test_partition_unknown_method():
    from adaqp_amd.graph import partition_all, tiny_ring_graph
    with pytest.raises(ValueError, match='unknown partition method'):
        partition_all(tiny_ring_graph(), 2, method='metis')


def test_nonbidirected_graph_rejected():
    """Asymmetric graphs must be refused at partition time (backward
    aggregation assumes the exact transpose)."""
    from adaqp_amd.graph import GlobalGraph, build_local_graph
    n = 6
    src = torch.tensor([0, 1, 2, 3, 4, 5, 0])   # one one-way edge 0->1
    dst = torch.tensor([0, 1, 2, 3, 4, 5, 1])
    g = GlobalGraph(n, src, dst, torch.zeros(n, 2), torch.zeros(n, dtype=torch.int64),
                    torch.ones(n, dtype=torch.bool), torch.zeros(n, dtype=torch.bool),
                    torch.zeros(n, dtype=torch.bool), 2, False)
    with pytest.raises(ValueError, match='not bidirected'):
        build_local_graph(g, torch.zeros(n, dtype=torch.int64), 0, 1)


def test_invalid_bit_width_rejected():
    from adaqp_amd.comm.buffers import _layout
    with pytest.raises(ValueError, match='invalid bit widths'):
        _layout([torch.tensor([2, 3, 8])], [torch.arange(3)], F=8)


def test_unknown_dataset_shape():
    from adaqp_amd.graph import synth_graph
    with pytest.raises(ValueError, match='unknown dataset shape'):
        synth_graph('citeseer', 2)


def test_results_table_empty_root(tmp_path):
    out = subprocess.run(
        [sys.executable, 'tools/results_table.py', '--root',
         str(tmp_path / 'nope')],
        cwd=REPO, capture_output=True, text=True, timeout=120)
    assert out.returncode != 0
    assert 'no metrics found' in (out.stderr + out.stdout)


def test_bench_refuses_unlaunched_multirank(tmp_path):
    """--gpus N>1 without torchrun must fail with guidance, not hang."""
    env = dict(os.environ)
    env.pop('WORLD_SIZE', None)
    env.pop('RANK', None)
    out = subprocess.run(
        [sys.executable, 'bench.py', '--gpus', '4', '--cpu',
         '--part-dir', str(tmp_path)],
        cwd=REPO, env=env, capture_output=True, text=True, timeout=300)
    assert out.returncode != 0
    assert 'torch.distributed.run' in (out.stderr + out.stdout)
