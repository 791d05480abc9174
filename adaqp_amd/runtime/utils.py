"""Epoch loop helpers: loss, metrics, gradient sync.

Reference parity: ``AdaQP/trainer/runtime_util.py`` (train_for_one_epoch,
val_test, aggregate_accuracy/aggregate_F1, average_gradients). Gradient
all-reduce is SUM with the loss pre-divided by the GLOBAL number of
training samples (``runtime_util.py:102``), so summed gradients are
correctly scaled — same convention here.
"""
from __future__ import annotations

from typing import Dict

import torch
import torch.nn.functional as F
from torch import Tensor

from ..comm.communicator import Communicator


def global_train_count(engine) -> Tensor:
    n = engine.graph.train_mask.sum().float().reshape(1)
    Communicator.ctx.all_reduce_sum(n)
    return n


def compute_loss(logits: Tensor, labels: Tensor, mask: Tensor,
                 multilabel: bool, global_count: Tensor) -> Tensor:
    if multilabel:
        raw = F.binary_cross_entropy_with_logits(
            logits[mask], labels[mask].float(), reduction='sum') / logits.shape[1]
    else:
        raw = F.cross_entropy(logits[mask], labels[mask], reduction='sum')
    return raw / global_count


def _autocast(engine):
    use = (engine.compute_dtype == torch.bfloat16
           and engine.device.type == 'cuda')
    return torch.autocast('cuda', dtype=torch.bfloat16, enabled=use) if use \
        else torch.autocast('cpu', enabled=False)


def train_epoch(engine, model, optimizer, global_count: Tensor,
                multilabel: bool) -> Tensor:
    comm = Communicator.ctx
    model.train()
    optimizer.zero_grad(set_to_none=False)
    with _autocast(engine):
        logits = model(engine, engine.graph.feats)
    logits = logits.float()
    loss = compute_loss(logits, engine.graph.labels, engine.graph.train_mask,
                        multilabel, global_count)
    loss.backward()
    with engine.timer.record('grad_reduce'):
        comm.flat_all_reduce_grads(model.parameters())
    optimizer.step()
    return loss.detach()


@torch.no_grad()
def evaluate(engine, model, multilabel: bool) -> Dict[str, float]:
    """Distributed train/val/test metrics: accuracy (single-label) or
    micro-F1 (multilabel), all-reduced counts (``runtime_util.py:139-197``)."""
    comm = Communicator.ctx
    model.eval()
    with _autocast(engine):
        logits = model(engine, engine.graph.feats)
    logits = logits.float()
    g = engine.graph
    out = {}
    if multilabel:
        pred = (logits > 0).float()
        for split, mask in (('train', g.train_mask), ('val', g.val_mask),
                            ('test', g.test_mask)):
            y = g.labels[mask]
            p = pred[mask]
            tp = (p * y).sum()
            counts = torch.stack([tp, p.sum(), y.sum()]).reshape(-1)
            comm.all_reduce_sum(counts)
            tp_g, ppos, apos = counts.tolist()
            prec = tp_g / max(ppos, 1.0)
            rec = tp_g / max(apos, 1.0)
            out[split] = 2 * prec * rec / max(prec + rec, 1e-12)
    else:
        pred = logits.argmax(dim=1)
        for split, mask in (('train', g.train_mask), ('val', g.val_mask),
                            ('test', g.test_mask)):
            correct = (pred[mask] == g.labels[mask]).sum().float()
            counts = torch.stack([correct, mask.sum().float()]).reshape(-1)
            comm.all_reduce_sum(counts)
            c, n = counts.tolist()
            out[split] = c / max(n, 1.0)
    return out
