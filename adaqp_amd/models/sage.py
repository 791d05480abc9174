"""DistSAGE: GraphSAGE with 'mean' / 'gcn' aggregators.

Reference parity: ``AdaQP/model/distSAGE.py:46-60`` — mean:
``fc_self(local) + fc_neigh(h_neigh)``; gcn: ``fc_neigh(h_neigh)`` where
h_neigh includes the self term (handled inside the aggregation op)."""
from __future__ import annotations

import torch
import torch.nn as nn
from torch import Tensor

from ..ops.dist_agg import dist_aggregate
from .common import FastLinear, fused_dual_linear, fused_dual_linear_ok


class DistSAGEConv(nn.Module):
    def __init__(self, in_dim: int, out_dim: int, layer: int,
                 aggregator_type: str = 'mean', use_bias: bool = True):
        super().__init__()
        assert aggregator_type in ('mean', 'gcn')
        self.layer = layer
        self.aggregator_type = aggregator_type
        self.fc_neigh = FastLinear(in_dim, out_dim, bias=use_bias)
        if aggregator_type == 'mean':
            self.fc_self = FastLinear(in_dim, out_dim, bias=use_bias)

    def reset_parameters(self):
        self.fc_neigh.reset_parameters()
        if self.aggregator_type == 'mean':
            self.fc_self.reset_parameters()

    def forward(self, engine, x: Tensor) -> Tensor:
        h_neigh = dist_aggregate(x, engine, self.layer, self.training)
        if self.aggregator_type == 'mean':
            x_self = x[:engine.graph.num_inner]
            if (engine.compute_dtype == torch.bfloat16
                    and fused_dual_linear_ok(x_self, h_neigh,
                                             self.fc_neigh.weight.shape[1])):
                return fused_dual_linear(x_self, h_neigh,
                                         self.fc_self, self.fc_neigh)
            return self.fc_self(x_self) + self.fc_neigh(h_neigh)
        return self.fc_neigh(h_neigh)


class DistSAGE(nn.Module):
    def __init__(self, in_dim: int, hidden_dim: int, out_dim: int,
                 num_layers: int = 3, dropout: float = 0.5,
                 use_norm: bool = True, aggregator_type: str = 'mean'):
        super().__init__()
        dims = [in_dim] + [hidden_dim] * (num_layers - 1) + [out_dim]
        self.convs = nn.ModuleList(
            [DistSAGEConv(dims[i], dims[i + 1], layer=i,
                          aggregator_type=aggregator_type)
             for i in range(num_layers)])
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
