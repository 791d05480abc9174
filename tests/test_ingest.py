"""Offline real-data ingestion: write a small
GraphSAINT-layout fixture to disk, load it, and run it through the FULL
pipeline — grow partitioning, LocalGraph build, engine forward."""
import json
import os
import subprocess
import sys

import numpy as np
import pytest
import torch

try:
    import scipy.sparse as sp
except ImportError:          # pragma: no cover
    sp = None

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def _write_fixture(d, n=60, multilabel=False, use_labels_npy=False, seed=0):
    rng = np.random.default_rng(seed)
    # random sparse graph, ~8 out-edges per node
    rows = np.repeat(np.arange(n), 8)
    cols = rng.integers(0, n, size=8 * n)
    data = np.ones_like(rows, dtype=np.float32)
    m = sp.csr_matrix((data, (rows, cols)), shape=(n, n))
    sp.save_npz(os.path.join(d, 'adj_full.npz'), m)
    np.save(os.path.join(d, 'feats.npy'),
            rng.standard_normal((n, 12)).astype(np.float32))
    if use_labels_npy:
        np.save(os.path.join(d, 'labels.npy'),
                rng.integers(0, 5, size=n).astype(np.int64))
    elif multilabel:
        cmap = {str(i): rng.integers(0, 2, size=4).tolist() for i in range(n)}
        json.dump(cmap, open(os.path.join(d, 'class_map.json'), 'w'))
    else:
        cmap = {str(i): int(rng.integers(0, 5)) for i in range(n)}
        json.dump(cmap, open(os.path.join(d, 'class_map.json'), 'w'))
    ids = rng.permutation(n)
    role = {'tr': ids[:n // 2].tolist(), 'va': ids[n // 2:3 * n // 4].tolist(),
            'te': ids[3 * n // 4:].tolist()}
    json.dump(role, open(os.path.join(d, 'role.json'), 'w'))


@pytest.mark.skipif(sp is None, reason='scipy required for fixture')
def test_load_graph_dir_roundtrip(tmp_path):
    from adaqp_amd.graph import load_graph_dir
    _write_fixture(str(tmp_path))
    g = load_graph_dir(str(tmp_path))
    assert g.num_nodes == 60
    assert g.feats.shape == (60, 12)
    assert not g.multilabel and g.num_classes == 5
    # masks partition the nodes
    assert bool((g.train_mask | g.val_mask | g.test_mask).all())
    assert not bool((g.train_mask & g.val_mask).any())
    # symmetrized + self-loops
    key = set(zip(g.src.tolist(), g.dst.tolist()))
    assert all((b, a) in key for a, b in key)
    assert all((i, i) in key for i in range(60))
    # train-fitted standardization: train rows ~zero-mean/unit-var
    tr = g.feats[g.train_mask]
    assert float(tr.mean(0).abs().max()) < 1e-4
    assert float((tr.std(0, unbiased=False) - 1).abs().max()) < 1e-3


@pytest.mark.skipif(sp is None, reason='scipy required for fixture')
def test_load_multilabel_and_labels_npy(tmp_path):
    from adaqp_amd.graph import load_graph_dir
    d1 = tmp_path / 'ml'
    d1.mkdir()
    _write_fixture(str(d1), multilabel=True)
    g = load_graph_dir(str(d1))
    assert g.multilabel and g.labels.shape == (60, 4)
    d2 = tmp_path / 'npy'
    d2.mkdir()
    _write_fixture(str(d2), use_labels_npy=True)
    g2 = load_graph_dir(str(d2))
    assert not g2.multilabel and g2.labels.dtype == torch.int64


@pytest.mark.skipif(sp is None, reason='scipy required for fixture')
def test_ingested_graph_full_pipeline(tmp_path):
    """disk -> load -> grow partition -> LocalGraph -> engine forward."""
    from adaqp_amd.graph import load_graph_dir, partition_all
    from adaqp_amd.runtime.engine import GraphEngine
    from adaqp_amd.helpers import RunMode, DistGNNType
    _write_fixture(str(tmp_path))
    g = load_graph_dir(str(tmp_path))
    parts = partition_all(g, 2, method='grow')
    assert sum(p.num_inner for p in parts) == g.num_nodes
    for lg in parts:
        lg.validate()


@pytest.mark.skipif(sp is None, reason='scipy required for fixture')
def test_partition_cli_raw_dir(tmp_path):
    raw = tmp_path / 'raw'
    raw.mkdir()
    _write_fixture(str(raw))
    out = subprocess.run(
        [sys.executable, 'graph_partition.py', '--dataset', 'yelp',
         '--raw_dir', str(raw), '--partition_size', '2',
         '--partition_dir', str(tmp_path / 'parts')],
        cwd=REPO, capture_output=True, text=True, timeout=300)
    assert out.returncode == 0, out.stderr[-1500:]
    assert 'saved 2 partitions' in out.stdout
    assert os.path.exists(tmp_path / 'parts' / 'yelp' / '2part' / 'part0.pt')
