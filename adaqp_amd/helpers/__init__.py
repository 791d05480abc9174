from .typing import DistGNNType, BitType, PropagationMode, RunMode, AssignScheme

__all__ = ['DistGNNType', 'BitType', 'PropagationMode', 'RunMode', 'AssignScheme']
