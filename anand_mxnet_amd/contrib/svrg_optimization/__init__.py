"""SVRG optimization (reference parity:
python/mxnet/contrib/svrg_optimization/ — SVRGModule / SVRGOptimizer)."""
from .svrg_module import SVRGModule
from .svrg_optimizer import _SVRGOptimizer
