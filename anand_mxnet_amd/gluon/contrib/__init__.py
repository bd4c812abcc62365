"""gluon.contrib (reference parity: SyncBatchNorm, estimator)."""
from . import nn
from . import estimator
