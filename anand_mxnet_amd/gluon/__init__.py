"""Gluon API (parity with python/mxnet/gluon)."""
from .parameter import Parameter, Constant, ParameterDict, \
    DeferredInitializationError
from .block import Block, HybridBlock, SymbolBlock
from .trainer import Trainer
from . import nn
from . import rnn
from . import loss
from . import data
from . import utils
from . import model_zoo
from . import contrib
