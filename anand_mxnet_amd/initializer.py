"""Weight initializers (parity with python/mxnet/initializer.py)."""
from __future__ import annotations

import math
import re

import numpy as np
import torch

from .ndarray.ndarray import NDArray

__all__ = ["Initializer", "Zero", "One", "Constant", "Uniform", "Normal",
           "Orthogonal", "Xavier", "MSRAPrelu", "Bilinear", "LSTMBias",
           "Mixed", "register", "create"]

_REGISTRY = {}


def register(cls):
    _REGISTRY[cls.__name__.lower()] = cls
    return cls


_ALIASES = {"zeros": "zero", "ones": "one", "gaussian": "normal"}


def create(name, **kwargs):
    if isinstance(name, Initializer):
        return name
    if name is None:
        return Uniform()
    key = name.lower()
    key = _ALIASES.get(key, key)
    return _REGISTRY[key](**kwargs)


class Initializer:
    """Base initializer.  Subclasses implement _init_weight."""

    def __init__(self, **kwargs):
        self._kwargs = kwargs

    def __call__(self, name, arr=None):
        if arr is None and not isinstance(name, str):
            name, arr = "weight", name
        self.init(name, arr)

    def init(self, name, arr):
        desc = name if isinstance(name, str) else str(name)
        if desc.endswith("bias"):
            self._init_bias(desc, arr)
        elif desc.endswith("gamma"):
            self._init_one(desc, arr)
        elif desc.endswith("beta"):
            self._init_zero(desc, arr)
        elif desc.endswith("running_mean") or desc.endswith("moving_mean"):
            self._init_zero(desc, arr)
        elif desc.endswith("running_var") or desc.endswith("moving_var"):
            self._init_one(desc, arr)
        else:
            self._init_weight(desc, arr)

    def _t(self, arr):
        return arr._t if isinstance(arr, NDArray) else arr

    def _fill(self, arr, fn):
        t = self._t(arr)
        with torch.no_grad():
            if t.dtype in (torch.bfloat16, torch.float16):
                t32 = torch.empty(t.shape, dtype=torch.float32, device=t.device)
                fn(t32)
                t.copy_(t32)
            else:
                fn(t)

    def _init_bias(self, name, arr):
        self._fill(arr, lambda t: t.zero_())

    def _init_zero(self, name, arr):
        self._fill(arr, lambda t: t.zero_())

    def _init_one(self, name, arr):
        self._fill(arr, lambda t: t.fill_(1.0))

    def _init_weight(self, name, arr):
        raise NotImplementedError

    def __repr__(self):
        return f"{self.__class__.__name__}({self._kwargs})"

    def dumps(self):
        import json
        return json.dumps([self.__class__.__name__.lower(), self._kwargs])


@register
class Zero(Initializer):
    def _init_weight(self, name, arr):
        self._fill(arr, lambda t: t.zero_())


@register
class One(Initializer):
    def _init_weight(self, name, arr):
        self._fill(arr, lambda t: t.fill_(1.0))


@register
class Constant(Initializer):
    def __init__(self, value=0.0):
        super().__init__(value=value)
        self.value = value

    def _init_weight(self, name, arr):
        self._fill(arr, lambda t: t.fill_(self.value))


@register
class Uniform(Initializer):
    def __init__(self, scale=0.07):
        super().__init__(scale=scale)
        self.scale = scale

    def _init_weight(self, name, arr):
        self._fill(arr, lambda t: t.uniform_(-self.scale, self.scale))


@register
class Normal(Initializer):
    def __init__(self, sigma=0.01):
        super().__init__(sigma=sigma)
        self.sigma = sigma

    def _init_weight(self, name, arr):
        self._fill(arr, lambda t: t.normal_(0, self.sigma))


@register
class Orthogonal(Initializer):
    def __init__(self, scale=1.414, rand_type="uniform"):
        super().__init__(scale=scale, rand_type=rand_type)
        self.scale = scale
        self.rand_type = rand_type

    def _init_weight(self, name, arr):
        t = self._t(arr)
        nout = t.shape[0]
        nin = int(np.prod(t.shape[1:]))
        if self.rand_type == "uniform":
            tmp = np.random.uniform(-1.0, 1.0, (nout, nin))
        else:
            tmp = np.random.normal(0.0, 1.0, (nout, nin))
        u, _, v = np.linalg.svd(tmp, full_matrices=False)
        q = u if u.shape == tmp.shape else v
        q = self.scale * q.reshape(t.shape)
        self._fill(arr, lambda d: d.copy_(torch.from_numpy(
            q.astype(np.float32)).to(d.device)))


@register
class Xavier(Initializer):
    def __init__(self, rnd_type="uniform", factor_type="avg", magnitude=3):
        super().__init__(rnd_type=rnd_type, factor_type=factor_type,
                         magnitude=magnitude)
        self.rnd_type = rnd_type
        self.factor_type = factor_type
        self.magnitude = float(magnitude)

    def _init_weight(self, name, arr):
        shape = tuple(self._t(arr).shape)
        hw_scale = 1.0
        if len(shape) < 2:
            raise ValueError(f"Xavier needs >=2d weight, got {shape} for {name}")
        if len(shape) > 2:
            hw_scale = float(np.prod(shape[2:]))
        fan_in, fan_out = shape[1] * hw_scale, shape[0] * hw_scale
        if self.factor_type == "avg":
            factor = (fan_in + fan_out) / 2.0
        elif self.factor_type == "in":
            factor = fan_in
        elif self.factor_type == "out":
            factor = fan_out
        else:
            raise ValueError("invalid factor_type")
        scale = math.sqrt(self.magnitude / factor)
        if self.rnd_type == "uniform":
            self._fill(arr, lambda t: t.uniform_(-scale, scale))
        elif self.rnd_type == "gaussian":
            self._fill(arr, lambda t: t.normal_(0, scale))
        else:
            raise ValueError("invalid rnd_type")


@register
class MSRAPrelu(Xavier):
    def __init__(self, factor_type="avg", slope=0.25):
        magnitude = 2.0 / (1 + slope ** 2)
        super().__init__("gaussian", factor_type, magnitude)
        self._kwargs = {"factor_type": factor_type, "slope": slope}


@register
class Bilinear(Initializer):
    def _init_weight(self, name, arr):
        t = self._t(arr)
        shape = t.shape
        weight = np.zeros(int(np.prod(shape)), dtype=np.float32)
        f = np.ceil(shape[3] / 2.0)
        c = (2 * f - 1 - f % 2) / (2.0 * f)
        for i in range(int(np.prod(shape))):
            x = i % shape[3]
            y = (i // shape[3]) % shape[2]
            weight[i] = (1 - abs(x / f - c)) * (1 - abs(y / f - c))
        w = weight.reshape(tuple(shape))
        self._fill(arr, lambda d: d.copy_(torch.from_numpy(w).to(d.device)))


@register
class LSTMBias(Initializer):
    def __init__(self, forget_bias=1.0):
        super().__init__(forget_bias=forget_bias)
        self.forget_bias = forget_bias

    def _init_weight(self, name, arr):
        t = self._t(arr)
        with torch.no_grad():
            t.zero_()
            n = t.shape[0] // 4
            t[n:2 * n] = self.forget_bias


class Mixed(Initializer):
    def __init__(self, patterns, initializers):
        super().__init__()
        self.map = list(zip([re.compile(p) for p in patterns], initializers))

    def init(self, name, arr):
        for prog, init_ in self.map:
            if prog.match(name):
                init_.init(name, arr)
                return
        raise ValueError(f"parameter {name} did not match any pattern")
