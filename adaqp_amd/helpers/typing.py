"""Core enums for the MI355X-native AdaQP framework.

Capability parity with the reference's ``AdaQP/helper/typing.py:1-26``
(DistGNNType / BitType / MessageType / ProprogationMode), re-designed:
MessageType p2p tags are gone because the RCCL data plane uses
``all_to_all_single`` (one fused collective) instead of tagged gloo
send/recv pairs.
"""
from enum import Enum


class DistGNNType(Enum):
    DistGCN = 0
    DistSAGE = 1


class BitType(Enum):
    FULL = 0    # fp32/bf16 full-precision boundary messages
    QUANT = 1   # mixed {2,4,8}-bit stochastic-quantized boundary messages


class PropagationMode(Enum):
    Forward = 0
    Backward = 1


class RunMode(Enum):
    """Run modes, same surface as the reference (``trainer.py:18-20``)."""
    Vanilla = 'Vanilla'     # fp messages, no overlap
    AdaQP = 'AdaQP'         # quantized messages + comp/comm overlap
    AdaQP_q = 'AdaQP-q'     # quantized messages only
    AdaQP_p = 'AdaQP-p'     # overlap only

    @property
    def bit_type(self) -> BitType:
        return BitType.QUANT if self in (RunMode.AdaQP, RunMode.AdaQP_q) else BitType.FULL

    @property
    def use_parallel(self) -> bool:
        return self in (RunMode.AdaQP, RunMode.AdaQP_p)


class AssignScheme(Enum):
    UNIFORM = 'uniform'
    RANDOM = 'random'
    ADAPTIVE = 'adaptive'
