"""Symbol attribute scoping (reference parity: python/mxnet/attribute.py —
AttrScope).  `with AttrScope(ctx_group='dev1'):` attaches the attributes to
every symbol created inside the scope (the reference's model-parallel
`group2ctx` placement rides on this)."""
from __future__ import annotations

import threading

_local = threading.local()


class AttrScope:
    """Attach attributes to all symbols created within the scope."""

    def __init__(self, **kwargs):
        for v in kwargs.values():
            if not isinstance(v, str):
                raise ValueError("AttrScope values must be strings")
        self._attr = kwargs
        self._old = None

    def get(self, attr):
        """Merge scope attrs under explicit per-symbol attrs."""
        if not self._attr:
            return attr
        merged = dict(self._attr)
        if attr:
            merged.update(attr)
        return merged

    def __enter__(self):
        self._old = current()
        merged = dict(self._old._attr)
        merged.update(self._attr)
        self._attr = merged
        _local.scope = self
        return self

    def __exit__(self, *exc):
        _local.scope = self._old
        return False


def current() -> AttrScope:
    sc = getattr(_local, "scope", None)
    if sc is None:
        sc = AttrScope()
        _local.scope = sc
    return sc
