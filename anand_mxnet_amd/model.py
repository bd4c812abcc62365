"""Checkpoint helpers (parity with python/mxnet/model.py:
save_checkpoint/load_checkpoint — prefix-symbol.json + prefix-%04d.params
with arg:/aux:-prefixed names)."""
from __future__ import annotations

from .serializer import save_ndarrays, load_ndarrays
from . import symbol as sym_mod

__all__ = ["save_checkpoint", "load_checkpoint", "BatchEndParam"]


def save_checkpoint(prefix, epoch, symbol, arg_params, aux_params,
                    remove_amp_cast=True):
    if symbol is not None:
        symbol.save(f"{prefix}-symbol.json")
    save_dict = {f"arg:{name}": v for name, v in arg_params.items()}
    save_dict.update({f"aux:{name}": v for name, v in aux_params.items()})
    param_name = f"{prefix}-{epoch:04d}.params"
    save_ndarrays(param_name, save_dict)


def load_checkpoint(prefix, epoch):
    import os
    symbol = None
    if os.path.exists(f"{prefix}-symbol.json"):
        symbol = sym_mod.load(f"{prefix}-symbol.json")
    save_dict = load_ndarrays(f"{prefix}-{epoch:04d}.params")
    arg_params, aux_params = {}, {}
    for k, v in save_dict.items():
        tp, name = k.split(":", 1)
        if tp == "arg":
            arg_params[name] = v
        elif tp == "aux":
            aux_params[name] = v
    return symbol, arg_params, aux_params


# re-export for callbacks
from .callback import BatchEndParam  # noqa: E402,F401
