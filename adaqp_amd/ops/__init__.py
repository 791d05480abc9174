from .dist_agg import DistAgg, dist_aggregate, fp_exchange, qt_exchange
from .kernels import spmm, SpmmView, mixed_quantize, mixed_dequantize, native, has_native

__all__ = ['DistAgg', 'dist_aggregate', 'fp_exchange', 'qt_exchange',
           'spmm', 'SpmmView', 'mixed_quantize', 'mixed_dequantize',
           'native', 'has_native']
