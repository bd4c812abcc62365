"""Device contexts (parity with mxnet.context: Context, cpu(), gpu(), ctx stack).

MI355X-native mapping: ``gpu(i)`` is HIP device i (torch.device('cuda', i) on
ROCm — torch's "cuda" namespace IS HIP on this stack).
"""
from __future__ import annotations

import threading
import torch

__all__ = ["Context", "cpu", "gpu", "cpu_pinned", "current_context", "num_gpus"]


class Context:
    devtype2str = {1: "cpu", 2: "gpu", 3: "cpu_pinned"}
    devstr2type = {v: k for k, v in devtype2str.items()}

    _default_ctx = threading.local()

    def __init__(self, device_type, device_id: int = 0):
        if isinstance(device_type, Context):
            self.device_type = device_type.device_type
            self.device_id = device_type.device_id
        else:
            self.device_type = Context.devstr2type[device_type] if isinstance(device_type, str) else device_type
            self.device_id = device_id

    @property
    def device_typestr(self) -> str:
        return Context.devtype2str[self.device_type]

    def torch_device(self) -> torch.device:
        if self.device_type == 2:
            return torch.device("cuda", self.device_id)
        return torch.device("cpu")

    @staticmethod
    def from_torch(dev: torch.device) -> "Context":
        if dev.type == "cuda":
            return Context("gpu", dev.index if dev.index is not None else 0)
        return Context("cpu", 0)

    def __eq__(self, other):
        return (isinstance(other, Context) and self.device_type == other.device_type
                and self.device_id == other.device_id)

    def __hash__(self):
        return hash((self.device_type, self.device_id))

    def __repr__(self):
        return f"{self.device_typestr}({self.device_id})"

    __str__ = __repr__

    def __enter__(self):
        if not hasattr(Context._default_ctx, "stack"):
            Context._default_ctx.stack = []
        Context._default_ctx.stack.append(self)
        return self

    def __exit__(self, *args):
        Context._default_ctx.stack.pop()

    def empty_cache(self):
        if self.device_type == 2 and torch.cuda.is_available():
            with torch.cuda.device(self.device_id):
                torch.cuda.empty_cache()


def current_context() -> Context:
    stack = getattr(Context._default_ctx, "stack", None)
    if stack:
        return stack[-1]
    return Context("cpu", 0)


def cpu(device_id: int = 0) -> Context:
    return Context("cpu", device_id)


def cpu_pinned(device_id: int = 0) -> Context:
    return Context("cpu_pinned", device_id)


def gpu(device_id: int = 0) -> Context:
    return Context("gpu", device_id)


def num_gpus() -> int:
    return torch.cuda.device_count() if torch.cuda.is_available() else 0
