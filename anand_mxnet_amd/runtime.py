"""Runtime feature detection (reference parity: python/mxnet/runtime.py —
`mx.runtime.Features`, `feature_list()`).  Thin view over libinfo."""
from __future__ import annotations

from . import libinfo
from .libinfo import Feature

__all__ = ["Features", "feature_list"]


def feature_list():
    return libinfo.features()


class Features(dict):
    """Dict-like: Features()['ROCM'].enabled / is_enabled('ROCM')."""

    def __init__(self):
        super().__init__([(f.name, f) for f in feature_list()])

    def is_enabled(self, name):
        feat = self.get(name)
        return bool(feat and feat.enabled)

    def __repr__(self):
        return "[" + ", ".join(repr(f) for f in self.values()) + "]"
