"""DistGCN: aggregate-then-transform GCN over partitioned graphs.

Reference parity: ``AdaQP/model/distGCN.py`` (DistGCNConv aggregates with
DistAggConv then applies the linear transform; stack uses
dropout -> LayerNorm -> ReLU between layers; Xavier init)."""
from __future__ import annotations

import torch
import torch.nn as nn
from torch import Tensor

from ..ops.dist_agg import dist_aggregate
from .common import FastLinear


class DistGCNConv(nn.Module):
    def __init__(self, in_dim: int, out_dim: int, layer: int, use_bias: bool = True):
        super().__init__()
        self.layer = layer
        self.linear = FastLinear(in_dim, out_dim, bias=use_bias)

    def reset_parameters(self):
        self.linear.reset_parameters()

    def forward(self, engine, x: Tensor) -> Tensor:
        rst = dist_aggregate(x, engine, self.layer, self.training)
        return self.linear(rst)


class DistGCN(nn.Module):
    def __init__(self, in_dim: int, hidden_dim: int, out_dim: int,
                 num_layers: int = 3, dropout: float = 0.5,
                 use_norm: bool = True):
        super().__init__()
        dims = [in_dim] + [hidden_dim] * (num_layers - 1) + [out_dim]
        self.convs = nn.ModuleList(
            [DistGCNConv(dims[i], dims[i + 1], layer=i) for i in range(num_layers)])
        self.norms = nn.ModuleList(
            [nn.LayerNorm(hidden_dim, elementwise_affine=True)
             for _ in range(num_layers - 1)]) if use_norm else None
        self.dropout = nn.Dropout(dropout)

    def forward(self, engine, feats: Tensor) -> Tensor:
        h = feats
        for i, conv in enumerate(self.convs):
            h = self.dropout(h) if i > 0 else h
            h = conv(engine, h)
            if i < len(self.convs) - 1:
                if self.norms is not None:
                    h = self.norms[i](h)
                h = torch.relu(h)
        return h
