"""Automatic symbol naming (reference parity: python/mxnet/name.py —
NameManager / Prefix).  `with name.Prefix('block_'):` prefixes every
auto-generated symbol name created inside the scope."""
from __future__ import annotations

import threading

_local = threading.local()


class NameManager:
    """Assigns unique names to symbols created without an explicit name."""

    def __init__(self):
        self._counter = {}
        self._old = None

    def get(self, name, hint):
        if name is not None:
            return name
        c = self._counter.get(hint, 0)
        self._counter[hint] = c + 1
        return f"{hint}{c}"

    def __enter__(self):
        self._old = current()
        _local.manager = self
        return self

    def __exit__(self, *exc):
        _local.manager = self._old
        return False


class Prefix(NameManager):
    """NameManager that prepends a fixed prefix to every generated name."""

    def __init__(self, prefix):
        super().__init__()
        self._prefix = prefix

    def get(self, name, hint):
        name = super().get(name, hint)
        return self._prefix + name


def current() -> NameManager:
    mgr = getattr(_local, "manager", None)
    if mgr is None:
        mgr = NameManager()
        _local.manager = mgr
    return mgr
