"""Build/runtime feature flags (reference parity: python/mxnet/libinfo.py
and src/libinfo.cc — `mx.runtime.Features`-style introspection).

MI355X-native: features reflect the ROCm/PyTorch substrate and which of
our HIP extensions actually loaded, so users can branch the way they did
on `mx.libinfo.features()`.
"""
from __future__ import annotations

__version__ = "1.0.0"


class Feature:
    def __init__(self, name, enabled):
        self.name = name
        self.enabled = enabled

    def __repr__(self):
        return f"✔ {self.name}" if self.enabled else f"✖ {self.name}"


def features():
    """Return the list of compiled/runtime features."""
    import torch
    feats = []
    hip = torch.version.hip is not None
    feats.append(Feature("ROCM", hip))
    feats.append(Feature("GPU", torch.cuda.is_available()))
    try:
        from . import _C  # noqa: F401
        native = True
    except Exception:
        native = False
    feats.append(Feature("NATIVE_KERNELS_GFX950", native))
    feats.append(Feature("RCCL", torch.distributed.is_nccl_available()))
    feats.append(Feature("GLOO", torch.distributed.is_gloo_available()))
    feats.append(Feature("BF16", True))
    feats.append(Feature("MX_FP8", native))
    feats.append(Feature("FLASH_ATTENTION", native))
    feats.append(Feature("HIPRTC", hip))
    feats.append(Feature("INT64_TENSOR_SIZE", True))
    return feats


def feature_dict():
    return {f.name: f.enabled for f in features()}
