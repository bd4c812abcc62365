"""gluon.contrib.nn (reference parity: SyncBatchNorm et al.)."""
from .basic_layers import SyncBatchNorm

__all__ = ["SyncBatchNorm"]
