"""CSR graph core for the MI355X-native framework.

The reference delegates all graph storage/aggregation to DGL
(``AdaQP/manager/graphEngine.py``, ``AdaQP/model/ops.py:30``); here the
graph is a plain CSR over torch tensors so the hot SpMM runs in our own
HIP kernel and the decomposition needed for comp/comm overlap is free:

Local node ordering per rank (same invariant as the reference's
``reorder_graph``, ``AdaQP/manager/conversion.py:56-90``):

    [0, C)      central  : inner nodes with NO remote in-neighbor
    [C, I)      marginal : inner nodes with >=1 remote in-neighbor
    [I, N)      remote   : halo nodes, grouped by owner rank, each
                           owner's block in the owner's send order

Because rows are sorted central-first, the central/marginal decomposition
used for overlap (`AdaQP/manager/conversion.py:114-172` builds two DGL
subgraphs + copy buffers) is just a ROW RANGE SPLIT of one CSR here: the
central SpMM covers rows [0, C) (all column indices < I, local data only)
and the marginal SpMM covers rows [C, I). No subgraph construction, no
src copy buffers.

Because remote nodes are stored in the exact order their owners send
them, the RCCL ``all_to_all_single`` output buffer IS the remote feature
block — the per-peer scatter of the reference (``op_util.py:168-170``)
disappears.

The framework targets bidirected (symmetric) graphs — every dataset the
reference ships is symmetrized + self-looped (``helper/partition.py:58-60``,
``conversion.py:28-32``). For a symmetric graph the backward pass reuses
the SAME CSR with swapped normalization vectors (exact transpose), so no
reverse graph is ever built (the reference's ``_set_bwd_graph``,
``graphEngine.py:135-147``).
"""
from __future__ import annotations

from dataclasses import dataclass, field
from typing import Dict, List, Optional

import torch
from torch import Tensor


def coo_to_csr(rows: Tensor, cols: Tensor, num_rows: int) -> tuple[Tensor, Tensor]:
    """Sort COO by row and build indptr. Returns (indptr[int64], indices[int64])."""
    order = torch.argsort(rows, stable=True)
    rows = rows[order]
    cols = cols[order]
    indptr = torch.zeros(num_rows + 1, dtype=torch.int64)
    counts = torch.bincount(rows, minlength=num_rows)
    indptr[1:] = torch.cumsum(counts, dim=0)
    return indptr, cols.contiguous()


@dataclass
class LocalGraph:
    """One rank's partition: in-edges of its inner nodes, reordered as above."""
    rank: int
    world_size: int
    num_central: int
    num_marginal: int
    num_nodes: int                 # inner + remote
    indptr: Tensor                 # int64 [num_inner+1] rows = inner nodes
    indices: Tensor                # int64 [E] cols in [0, num_nodes)
    in_deg: Tensor                 # float32 [num_nodes] GLOBAL in-degrees
    out_deg: Tensor                # float32 [num_nodes] GLOBAL out-degrees
    # boundary exchange structure
    send_idx: Dict[int, Tensor]    # peer -> int64 local inner idx to send (in send order)
    recv_splits: List[int]         # len world: #remote nodes owned by each peer
    # mapping local -> global node id (for bookkeeping/save-load)
    local_to_global: Tensor        # int64 [num_nodes]
    # training data for inner nodes
    feats: Optional[Tensor] = None
    labels: Optional[Tensor] = None
    train_mask: Optional[Tensor] = None
    val_mask: Optional[Tensor] = None
    test_mask: Optional[Tensor] = None
    # derived (filled in __post_init__)
    total_send_idx: Tensor = field(init=False)
    send_splits: List[int] = field(init=False)

    def __post_init__(self):
        parts = []
        splits = []
        for p in range(self.world_size):
            idx = self.send_idx.get(p)
            if p == self.rank or idx is None or idx.numel() == 0:
                splits.append(0)
            else:
                parts.append(idx)
                splits.append(int(idx.numel()))
        self.total_send_idx = (torch.cat(parts) if parts
                               else torch.empty(0, dtype=torch.int64))
        self.send_splits = splits
        assert len(self.recv_splits) == self.world_size
        assert sum(self.recv_splits) == self.num_remote

    # ---- basic sizes ----
    @property
    def num_inner(self) -> int:
        return self.num_central + self.num_marginal

    @property
    def num_remote(self) -> int:
        return self.num_nodes - self.num_inner

    @property
    def num_edges(self) -> int:
        return int(self.indices.numel())

    @property
    def num_send(self) -> int:
        return int(self.total_send_idx.numel())

    # ---- decomposition views (zero-copy row-range splits) ----
    def central_view(self) -> tuple[Tensor, Tensor, int]:
        """(indptr, indices, base) covering rows [0, C)."""
        e = self.indptr[self.num_central]
        return self.indptr[:self.num_central + 1], self.indices[:e], 0

    def marginal_view(self) -> tuple[Tensor, Tensor, int]:
        """(indptr, indices, base) covering rows [C, I); indptr re-based."""
        s = self.indptr[self.num_central]
        ptr = self.indptr[self.num_central:] - s
        return ptr, self.indices[s:], int(self.num_central)

    def validate(self) -> None:
        C, I, N = self.num_central, self.num_inner, self.num_nodes
        assert 0 <= C <= I <= N
        assert self.indptr.numel() == I + 1
        assert int(self.indptr[-1]) == self.indices.numel()
        if self.indices.numel():
            assert int(self.indices.max()) < N
            assert int(self.indices.min()) >= 0
        # central rows must only reference local (inner) columns
        e = int(self.indptr[C])
        if e:
            assert int(self.indices[:e].max()) < I, \
                'central rows reference remote columns: reorder is broken'
        assert self.in_deg.numel() == N and self.out_deg.numel() == N
        assert self.local_to_global.numel() == N
        for p, idx in self.send_idx.items():
            assert p != self.rank
            if idx.numel():
                assert int(idx.max()) < I, 'send_idx must point at inner nodes'

    def to(self, device) -> 'LocalGraph':
        """Move all tensors to device in place; returns self."""
        for name in ('indptr', 'indices', 'in_deg', 'out_deg', 'local_to_global',
                     'total_send_idx', 'feats', 'labels', 'train_mask',
                     'val_mask', 'test_mask'):
            t = getattr(self, name)
            if t is not None:
                setattr(self, name, t.to(device))
        self.send_idx = {p: v.to(device) for p, v in self.send_idx.items()}
        return self


def pad_feat_dim(graph: 'LocalGraph', multiple: int = 8) -> int:
    """Zero-pad the feature dim to a multiple of ``multiple`` so every
    SpMM row-base load is wide-vector aligned (602 -> 608 etc.). Zero
    columns are inert: their weights receive zero gradient. Returns the
    padded dim."""
    if graph.feats is None:
        return 0
    F = graph.feats.shape[1]
    pad = (-F) % multiple
    if pad:
        graph.feats = torch.nn.functional.pad(graph.feats, (0, pad))
    return F + pad
