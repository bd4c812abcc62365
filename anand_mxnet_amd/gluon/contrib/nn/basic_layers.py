"""Contrib layers (reference parity: gluon/contrib/nn/basic_layers.py).

SyncBatchNorm: cross-rank batch normalization — batch statistics are
all-reduced over torch.distributed (RCCL on GPU, gloo on CPU) so every
rank normalizes with the GLOBAL batch mean/var.  Reference used a
key-matched allreduce through the KVStore; here the stats ride one small
fused [2C+1] collective per layer.
"""
from __future__ import annotations

import torch
import torch.distributed as dist

from ....ndarray.ndarray import NDArray
from ...nn.basic_layers import BatchNorm
from .... import autograd as _ag


class SyncBatchNorm(BatchNorm):
    def __init__(self, in_channels=0, num_devices=None, momentum=0.9,
                 epsilon=1e-5, center=True, scale=True,
                 use_global_stats=False, beta_initializer="zeros",
                 gamma_initializer="ones",
                 running_mean_initializer="zeros",
                 running_variance_initializer="ones", **kwargs):
        super().__init__(axis=1, momentum=momentum, epsilon=epsilon,
                         center=center, scale=scale,
                         use_global_stats=use_global_stats,
                         beta_initializer=beta_initializer,
                         gamma_initializer=gamma_initializer,
                         running_mean_initializer=running_mean_initializer,
                         running_variance_initializer=running_variance_initializer,
                         in_channels=in_channels, **kwargs)
        self._eps = epsilon
        self._momentum = momentum

    def hybrid_forward(self, F, x, addend=None, gamma=None, beta=None,
                       running_mean=None, running_var=None):
        t = x._t if isinstance(x, NDArray) else x
        g = gamma._t if isinstance(gamma, NDArray) else gamma
        b = beta._t if isinstance(beta, NDArray) else beta
        rm = running_mean._t if isinstance(running_mean, NDArray) \
            else running_mean
        rv = running_var._t if isinstance(running_var, NDArray) \
            else running_var
        training = _ag.is_training() and not self._kwargs.get(
            "use_global_stats", False)
        synced = dist.is_available() and dist.is_initialized() and \
            dist.get_world_size() > 1
        with torch.set_grad_enabled(_ag.is_recording()):
            if not (training and synced):
                out = torch.nn.functional.batch_norm(
                    t, rm, rv, g, b, training, 1.0 - self._momentum,
                    self._eps)
                return NDArray(out)
            # global stats: one fused all-reduce of [sum, sumsq, count]
            dims = [d for d in range(t.dim()) if d != 1]
            cnt = t.numel() // t.shape[1]
            s = t.float().sum(dim=dims)
            ss = (t.float() * t.float()).sum(dim=dims)
            C = s.numel()
            pack = torch.empty(2 * C + 1, device=t.device,
                               dtype=torch.float32)
            pack[:C] = s.detach()
            pack[C:2 * C] = ss.detach()
            pack[2 * C] = float(cnt)
            dist.all_reduce(pack)
            tot = pack[2 * C]
            mean = pack[:C] / tot
            var = pack[C:2 * C] / tot - mean * mean
            with torch.no_grad():
                m = self._momentum
                rm.mul_(m).add_(mean * (1 - m))
                unb = var * tot / (tot - 1) if tot > 1 else var
                rv.mul_(m).add_(unb * (1 - m))
            shape = [1, C] + [1] * (t.dim() - 2)
            xh = (t - mean.reshape(shape).to(t.dtype)) \
                / torch.sqrt(var.reshape(shape) + self._eps).to(t.dtype)
            out = xh * g.reshape(shape).to(t.dtype) \
                + b.reshape(shape).to(t.dtype)
            return NDArray(out)
