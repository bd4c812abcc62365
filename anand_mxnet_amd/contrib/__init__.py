from . import amp
from . import quantization
from . import text
