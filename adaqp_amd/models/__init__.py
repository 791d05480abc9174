from .gcn import DistGCN, DistGCNConv
from .sage import DistSAGE, DistSAGEConv

__all__ = ['DistGCN', 'DistGCNConv', 'DistSAGE', 'DistSAGEConv']
