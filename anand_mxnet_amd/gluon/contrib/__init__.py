"""gluon.contrib (parity subset: SyncBatchNorm via RCCL all-reduce —
reference src/operator/contrib/sync_batch_norm-inl.h used an intra-process
barrier + CPU reduce; here cross-rank (sum, sqsum, count) all-reduce)."""
from __future__ import annotations

import torch
import torch.distributed as dist

from ..nn.basic_layers import BatchNorm

__all__ = ["SyncBatchNorm"]


class SyncBatchNorm(BatchNorm):
    """BatchNorm with cross-rank statistics (training mode, dist initialized)."""

    def __init__(self, in_channels=0, num_devices=None, momentum=0.9,
                 epsilon=1e-5, center=True, scale=True,
                 use_global_stats=False, beta_initializer="zeros",
                 gamma_initializer="ones", running_mean_initializer="zeros",
                 running_variance_initializer="ones", **kwargs):
        super().__init__(axis=1, momentum=momentum, epsilon=epsilon,
                         center=center, scale=scale,
                         use_global_stats=use_global_stats,
                         beta_initializer=beta_initializer,
                         gamma_initializer=gamma_initializer,
                         running_mean_initializer=running_mean_initializer,
                         running_variance_initializer=running_variance_initializer,
                         in_channels=in_channels, **kwargs)

    def hybrid_forward(self, F, x, gamma, beta, running_mean, running_var):
        from ... import autograd as ag
        if not (ag.is_training() and dist.is_initialized()
                and dist.get_world_size() > 1):
            return super().hybrid_forward(F, x, gamma, beta, running_mean,
                                          running_var)
        # cross-rank mean/var: all-reduce (sum, sqsum, count) then normalize
        t = x._t
        dims = [d for d in range(t.dim()) if d != 1]
        n_local = t.numel() // t.shape[1]
        s = t.float().sum(dim=dims)
        ss = (t.float() ** 2).sum(dim=dims)
        stats = torch.cat([s, ss, torch.full((1,), float(n_local),
                                             device=t.device)])
        dist.all_reduce(stats)
        C = t.shape[1]
        n = stats[-1]
        mean = stats[:C] / n
        var = stats[C:2 * C] / n - mean ** 2
        with torch.no_grad():
            m = self._kwargs["momentum"]
            running_mean._t.mul_(m).add_((1 - m) * mean)
            running_var._t.mul_(m).add_((1 - m) * var)
        shape = [1, C] + [1] * (t.dim() - 2)
        eps = self._kwargs["eps"]
        xn = (t.float() - mean.reshape(shape)) * torch.rsqrt(
            var.reshape(shape) + eps)
        out = xn * gamma._t.reshape(shape) + beta._t.reshape(shape)
        from ...ndarray.ndarray import NDArray
        return NDArray(out.to(t.dtype))
