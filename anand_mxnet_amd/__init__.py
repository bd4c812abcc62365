"""anand_mxnet_amd — a brand-new MI355X-native deep-learning framework with
the capabilities of the reference (anandj91/anand-mxnet, an Apache MXNet 1.x
fork): mxnet.nd / Gluon / Module public API, the .params checkpoint format,
async stream-ordered execution, and KVStore data parallelism re-designed as
RCCL collectives over xGMI.

Import as ``import anand_mxnet_amd as mx``.
"""

__version__ = "0.1.0"

from . import base
from .base import MXNetError
from . import context
from .context import Context, cpu, gpu, cpu_pinned, current_context, num_gpus
from . import engine
from . import autograd
from . import ndarray
from . import ndarray as nd
from . import random
from . import random as random_mod
from . import initializer
from . import initializer as init
from . import optimizer
from . import lr_scheduler
from . import metric
from . import kvstore
from . import kvstore as kv
from . import gluon
from . import module
from . import module as mod
from . import model
from .model import save_checkpoint, load_checkpoint
from . import io
from . import recordio
from . import callback
from . import profiler
from . import symbol
from . import symbol as sym
from . import rtc
from . import contrib
from . import monitor
from . import operator
from . import image
from . import visualization
from . import libinfo
from . import runtime
from . import attribute
from . import name
from . import kvstore_server
from . import test_utils
from . import util


def waitall():
    nd.waitall()
