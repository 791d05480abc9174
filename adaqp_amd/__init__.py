import torch as _torch  # the _C HIP extension links against libtorch

__version__ = '0.1.0'
