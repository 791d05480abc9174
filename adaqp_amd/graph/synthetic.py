"""Synthetic graph generation.

There is no network in the target environment (BASELINE.json: benchmarks
run on "synthetic graphs of the named shape with random-init weights"),
so each reference dataset is mirrored by a synthetic generator producing
a graph of the same node count / edge count / feature dim / class count
(shapes from the reference's configs + public dataset stats,
``/root/reference/AdaQP/config/*.yaml``).

Structure: nodes are split into ``num_parts`` contiguous ranges with
planted partition locality — a fraction ``cut_frac`` of each node's
in-edges come from other parts (mimicking a METIS partition's edge cut,
``/root/reference/AdaQP/helper/partition.py:70-72``), the rest from its
own part with a power-law-ish source distribution (hub nodes at the low
end of each range). The graph is symmetrized (both directions present)
and self-loops are added, matching the reference's preprocessing
(``helper/partition.py:58-60``). Generation is deterministic in
``seed`` so every rank builds the identical global graph.
"""
from __future__ import annotations

from dataclasses import dataclass
from typing import Dict, Optional, Tuple

import torch
from torch import Tensor

# (num_nodes, num_directed_edges_before_sym, feat_dim, num_classes, multilabel)
DATASET_SHAPES: Dict[str, Tuple[int, int, int, int, bool]] = {
    'reddit':         (232_965,   114_615_892, 602, 41,  False),
    'yelp':           (716_847,   13_954_819,  300, 100, True),
    'ogbn-products':  (2_449_029, 123_718_280, 100, 47,  False),
    'amazonProducts': (1_569_960, 264_339_468, 200, 107, True),
}


@dataclass
class GlobalGraph:
    num_nodes: int
    src: Tensor            # int64 [E] (symmetric: both directions present)
    dst: Tensor            # int64 [E]
    feats: Tensor          # float32 [N, F]
    labels: Tensor         # int64 [N] or float32 [N, C] (multilabel)
    train_mask: Tensor     # bool [N]
    val_mask: Tensor
    test_mask: Tensor
    num_classes: int
    multilabel: bool

    @property
    def num_edges(self) -> int:
        return int(self.src.numel())


def _dedup_edges(src: Tensor, dst: Tensor, n: int) -> Tuple[Tensor, Tensor]:
    """Deduplicate (src,dst) pairs. The unique/sort over E*2 int64 keys is
    the generator's hot spot (amazonProducts: ~5*10^8 keys, minutes on
    CPU) — run it on the GPU when one is present (seconds; the result is
    deterministic either way)."""
    dev = 'cuda' if torch.cuda.is_available() else 'cpu'
    key = (src.to(dev) * n + dst.to(dev)).contiguous()
    key = torch.unique(key)
    out = key // n, key % n
    return out[0].cpu(), out[1].cpu()


def synth_graph(name: str, num_parts: int, seed: int = 0, cut_frac: float = 0.10,
                scale: float = 1.0, feat_dim: Optional[int] = None,
                teacher_labels: bool = True) -> GlobalGraph:
    """Build a synthetic graph shaped like dataset ``name``.

    ``scale`` < 1 shrinks nodes+edges proportionally (for quick tests).
    """
    if name not in DATASET_SHAPES:
        raise ValueError(f'unknown dataset shape {name}; options: {list(DATASET_SHAPES)}')
    n, e, f, c, multilabel = DATASET_SHAPES[name]
    n = max(int(n * scale), 8 * max(num_parts, 1))
    e = max(int(e * scale), 4 * n)
    if feat_dim is not None:
        f = feat_dim
    return random_partitioned_graph(n, e, f, c, num_parts, seed=seed,
                                    cut_frac=cut_frac, multilabel=multilabel,
                                    teacher_labels=teacher_labels)


def _teacher_labels(src: Tensor, dst: Tensor, feats: Tensor, num_classes: int,
                    gen: torch.Generator, multilabel: bool = False) -> Tensor:
    """Labels from a random 1-hop mean-aggregation linear teacher, so synthetic
    graphs are LEARNABLE (accuracy comparisons Vanilla vs AdaQP are
    meaningful — the reference uses real labeled datasets). Multilabel:
    each class is positive where its teacher logit is in the top decile
    (~10% positive rate, yelp-like)."""
    n, f = feats.shape
    dev = 'cuda' if torch.cuda.is_available() else 'cpu'
    fd = feats.to(dev)
    sd, dd = src.to(dev), dst.to(dev)
    deg = torch.bincount(dd, minlength=n).float().clamp(min=1)
    agg = torch.zeros_like(fd)
    agg.index_add_(0, dd, fd[sd])
    h = agg / deg[:, None]
    wout = torch.randn(f, num_classes, generator=gen).to(dev)
    logits = h @ wout
    if multilabel:
        # per-class 90th percentile via sort (torch.quantile caps at
        # 16M elements; amazonProducts logits are 168M)
        k = min(int(0.9 * n), n - 1)
        thr = logits.float().sort(dim=0).values[k:k + 1]
        return (logits > thr).float().cpu()
    return logits.argmax(dim=1).cpu()


def random_partitioned_graph(num_nodes: int, num_edges: int, feat_dim: int,
                             num_classes: int, num_parts: int, *, seed: int = 0,
                             cut_frac: float = 0.10, multilabel: bool = False,
                             alpha: float = 2.0,
                             teacher_labels: bool = False) -> GlobalGraph:
    g = torch.Generator().manual_seed(seed)
    P = max(num_parts, 1)
    bounds = torch.linspace(0, num_nodes, P + 1, dtype=torch.int64)
    # undirected pair budget: symmetrization roughly doubles, self-loops add n
    m = max(num_edges // 2, num_nodes)

    # dst uniform over all nodes; src power-law *within* a part
    dst = torch.randint(0, num_nodes, (m,), generator=g)
    dpart = torch.bucketize(dst, bounds[1:-1], right=True)
    cross = torch.rand(m, generator=g) < cut_frac
    # pick source part: own part, or a uniformly random other part for cut edges
    spart = dpart.clone()
    if P > 1 and cross.any():
        shift = torch.randint(1, P, (int(cross.sum()),), generator=g)
        spart[cross] = (dpart[cross] + shift) % P
    lo = bounds[spart]
    size = (bounds[spart + 1] - lo).to(torch.float64)
    u = torch.rand(m, generator=g, dtype=torch.float64)
    src = lo + (u.pow(alpha) * size).to(torch.int64)
    src = torch.minimum(src, bounds[spart + 1] - 1)

    # symmetrize + self loops + dedup
    s = torch.cat([src, dst, torch.arange(num_nodes)])
    d = torch.cat([dst, src, torch.arange(num_nodes)])
    keep = torch.ones_like(s, dtype=torch.bool)
    s, d = _dedup_edges(s[keep], d[keep], num_nodes)

    feats = torch.randn(num_nodes, feat_dim, generator=g)
    if multilabel and teacher_labels:
        labels = _teacher_labels(s, d, feats, num_classes, g, multilabel=True)
    elif multilabel:
        labels = (torch.rand(num_nodes, num_classes, generator=g) < 0.1).float()
    elif teacher_labels:
        labels = _teacher_labels(s, d, feats, num_classes, g)
    else:
        labels = torch.randint(0, num_classes, (num_nodes,), generator=g)
    r = torch.rand(num_nodes, generator=g)
    train_mask = r < 0.66
    val_mask = (r >= 0.66) & (r < 0.8)
    test_mask = r >= 0.8
    return GlobalGraph(num_nodes, s, d, feats, labels, train_mask, val_mask,
                       test_mask, num_classes, multilabel)


def tiny_ring_graph(num_nodes: int = 16, feat_dim: int = 4, num_classes: int = 3,
                    seed: int = 0, extra_edges: int = 16) -> GlobalGraph:
    """Small deterministic symmetric graph for unit tests."""
    g = torch.Generator().manual_seed(seed)
    i = torch.arange(num_nodes)
    s0, d0 = i, (i + 1) % num_nodes
    es = torch.randint(0, num_nodes, (extra_edges,), generator=g)
    ed = torch.randint(0, num_nodes, (extra_edges,), generator=g)
    s = torch.cat([s0, d0, es, ed, i])
    d = torch.cat([d0, s0, ed, es, i])
    s, d = _dedup_edges(s, d, num_nodes)
    feats = torch.randn(num_nodes, feat_dim, generator=g)
    labels = torch.randint(0, num_classes, (num_nodes,), generator=g)
    r = torch.rand(num_nodes, generator=g)
    return GlobalGraph(num_nodes, s, d, feats, labels, r < 0.5,
                       (r >= 0.5) & (r < 0.75), r >= 0.75, num_classes, False)
