from .communicator import Communicator
from .buffers import (KeyPlan, SidePlan, BITS_SET, bytes_per_node,
                      build_key_plan, uniform_bits, exchange_bits)

__all__ = ['Communicator', 'KeyPlan', 'SidePlan', 'BITS_SET', 'bytes_per_node',
           'build_key_plan', 'uniform_bits', 'exchange_bits']
