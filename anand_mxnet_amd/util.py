"""Misc utilities (parity subset of python/mxnet/util.py)."""
from __future__ import annotations

import functools
import inspect

__all__ = ["use_np_shape", "makedirs", "get_gpu_count", "getenv", "setenv"]


def makedirs(d):
    import os
    os.makedirs(d, exist_ok=True)


def get_gpu_count():
    from .context import num_gpus
    return num_gpus()


def getenv(name):
    import os
    return os.environ.get(name)


def setenv(name, value):
    import os
    os.environ[name] = value


def use_np_shape(func):
    @functools.wraps(func)
    def wrapper(*args, **kwargs):
        return func(*args, **kwargs)
    return wrapper
