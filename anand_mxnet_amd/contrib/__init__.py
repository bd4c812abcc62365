from . import amp
from . import quantization
