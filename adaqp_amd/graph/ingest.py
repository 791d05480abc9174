"""Offline real-data ingestion: load a graph dataset from disk.

The reference loads Reddit/OGB via framework downloaders and yelp /
amazonProducts from on-disk scipy-CSR + json files
(``/root/reference/AdaQP/helper/dataset.py:123-161``). This environment
has no network, so the synthetic generator covers the benchmark regime —
but the FRAMEWORK still needs an ingestion path for real data. This
module reads the same GraphSAINT-style layout the reference consumes:

    <dir>/adj_full.npz      scipy CSR (shape [N, N]); row i's columns
                            are the out-neighbors of node i
    <dir>/feats.npy         float [N, F]
    <dir>/class_map.json    {node_id: class} or {node_id: [multilabel]}
                            (labels.npy [N] or [N, C] is also accepted)
    <dir>/role.json         {'tr': [...], 'va': [...], 'te': [...]}

and returns a ``GlobalGraph`` ready for ``partition_all`` /
``build_local_graph``. Feature standardization fits on train rows only,
matching the reference (``dataset.py:154-158``); the graph is
symmetrized and self-looped like every reference dataset
(``helper/partition.py:58-60``).
"""
from __future__ import annotations

import json
import os
from typing import Optional

import numpy as np
import torch

from .synthetic import GlobalGraph, _dedup_edges


def _load_adj_coo(path: str):
    """Read a scipy ``save_npz`` CSR/COO file without importing scipy at
    module scope (npz keys: data/indices/indptr/shape or row/col)."""
    with np.load(path) as z:
        if 'indptr' in z:
            indptr = torch.from_numpy(z['indptr'].astype(np.int64))
            indices = torch.from_numpy(z['indices'].astype(np.int64))
            n = int(z['shape'][0])
            counts = indptr[1:] - indptr[:-1]
            src = torch.repeat_interleave(torch.arange(n), counts)
            return src, indices, n
        if 'row' in z:   # COO save_npz
            src = torch.from_numpy(z['row'].astype(np.int64))
            dst = torch.from_numpy(z['col'].astype(np.int64))
            return src, dst, int(z['shape'][0])
    raise ValueError(f'{path}: not a scipy-saved sparse matrix '
                     '(expected indptr/indices/shape or row/col keys)')


def load_graph_dir(raw_dir: str, standardize: bool = True,
                   symmetrize: bool = True,
                   num_classes: Optional[int] = None) -> GlobalGraph:
    """Load a GraphSAINT-layout dataset directory into a GlobalGraph."""
    adj = os.path.join(raw_dir, 'adj_full.npz')
    if not os.path.exists(adj):
        raise FileNotFoundError(f'{adj} not found')
    src, dst, n = _load_adj_coo(adj)

    feats = torch.from_numpy(
        np.load(os.path.join(raw_dir, 'feats.npy'))).float()
    if feats.shape[0] != n:
        raise ValueError(f'feats rows {feats.shape[0]} != adj nodes {n}')

    # labels: labels.npy beats class_map.json when both exist
    lab_npy = os.path.join(raw_dir, 'labels.npy')
    if os.path.exists(lab_npy):
        raw = np.load(lab_npy)
        multilabel = raw.ndim == 2
        labels = (torch.from_numpy(raw).float() if multilabel
                  else torch.from_numpy(raw.astype(np.int64)))
    else:
        with open(os.path.join(raw_dir, 'class_map.json')) as f:
            cmap = json.load(f)
        vals = [cmap[k] for k in sorted(cmap, key=lambda s: int(s))]
        multilabel = isinstance(vals[0], (list, tuple))
        labels = (torch.tensor(vals, dtype=torch.float32) if multilabel
                  else torch.tensor(vals, dtype=torch.int64))
    if len(labels) != n:
        raise ValueError(f'labels rows {len(labels)} != adj nodes {n}')
    if num_classes is None:
        num_classes = (labels.shape[1] if multilabel
                       else int(labels.max()) + 1)

    with open(os.path.join(raw_dir, 'role.json')) as f:
        role = json.load(f)
    train_mask = torch.zeros(n, dtype=torch.bool)
    val_mask = torch.zeros(n, dtype=torch.bool)
    test_mask = torch.zeros(n, dtype=torch.bool)
    train_mask[torch.tensor(role['tr'], dtype=torch.int64)] = True
    val_mask[torch.tensor(role['va'], dtype=torch.int64)] = True
    test_mask[torch.tensor(role['te'], dtype=torch.int64)] = True

    if standardize:
        # fit on train rows only (reference dataset.py:154-158)
        tr = feats[train_mask]
        mu = tr.mean(dim=0)
        sd = tr.std(dim=0, unbiased=False).clamp(min=1e-12)
        feats = (feats - mu) / sd

    if symmetrize:
        i = torch.arange(n)
        s = torch.cat([src, dst, i])
        d = torch.cat([dst, src, i])
        src, dst = _dedup_edges(s, d, n)

    return GlobalGraph(n, src, dst, feats, labels, train_mask, val_mask,
                       test_mask, num_classes, multilabel)
