"""Optimizer update ops (reference: src/operator/optimizer_op.{cc,cu} —
sgd_update, sgd_mom_update, mp_sgd_*, adam_update, lamb_update_phase1/2,
multi_* fused variants).

All updates are in-place on weight/state tensors and run outside autograd.
GPU path: hand-written multi-tensor HIP kernels (ops.hip.optim) — one launch
per bucket of tensors, bf16 weights with fp32 master copies for the mp_*
variants.  CPU path: torch reference arithmetic (the test oracle).
"""
from __future__ import annotations

import torch

from .registry import register, gpu_impl, has_gpu_impl


def _maybe_hip(name, *tensors):
    t = tensors[0]
    return t.is_cuda and has_gpu_impl(name)


@register("sgd_update")
def sgd_update(weight, grad, lr, wd=0.0, rescale_grad=1.0, clip_gradient=-1.0,
               lazy_update=False, out=None):
    with torch.no_grad():
        if _maybe_hip("sgd_update", weight):
            gpu_impl("sgd_update")(weight, grad, lr, wd, rescale_grad,
                                   clip_gradient)
            return weight
        g = grad.float() * rescale_grad
        if clip_gradient > 0:
            g = g.clamp(-clip_gradient, clip_gradient)
        g = g + wd * weight.float()
        weight.sub_((lr * g).to(weight.dtype))
    return weight


@register("sgd_mom_update")
def sgd_mom_update(weight, grad, mom, lr, momentum=0.0, wd=0.0,
                   rescale_grad=1.0, clip_gradient=-1.0, lazy_update=False,
                   out=None):
    with torch.no_grad():
        if _maybe_hip("sgd_mom_update", weight):
            gpu_impl("sgd_mom_update")(weight, grad, mom, lr, momentum, wd,
                                       rescale_grad, clip_gradient)
            return weight
        g = grad.to(mom.dtype) * rescale_grad
        if clip_gradient > 0:
            g = g.clamp(-clip_gradient, clip_gradient)
        g = g + wd * weight.to(mom.dtype)
        mom.mul_(momentum).sub_(lr * g)
        weight.add_(mom.to(weight.dtype))
    return weight


@register("mp_sgd_mom_update")
def mp_sgd_mom_update(weight, grad, mom, weight32, lr, momentum=0.0, wd=0.0,
                      rescale_grad=1.0, clip_gradient=-1.0, out=None):
    """Multi-precision: bf16/fp16 weight, fp32 master weight + momentum."""
    with torch.no_grad():
        if _maybe_hip("mp_sgd_mom_update", weight):
            gpu_impl("mp_sgd_mom_update")(weight, grad, mom, weight32, lr,
                                          momentum, wd, rescale_grad,
                                          clip_gradient)
            return weight
        g = grad.float() * rescale_grad
        if clip_gradient > 0:
            g = g.clamp(-clip_gradient, clip_gradient)
        g = g + wd * weight32
        mom.mul_(momentum).sub_(lr * g)
        weight32.add_(mom)
        weight.copy_(weight32.to(weight.dtype))
    return weight


@register("adam_update")
def adam_update(weight, grad, mean, var, lr, beta1=0.9, beta2=0.999,
                epsilon=1e-8, wd=0.0, rescale_grad=1.0, clip_gradient=-1.0,
                lazy_update=False, out=None):
    with torch.no_grad():
        if _maybe_hip("adam_update", weight):
            gpu_impl("adam_update")(weight, grad, mean, var, lr, beta1, beta2,
                                    epsilon, wd, rescale_grad, clip_gradient)
            return weight
        g = grad.to(mean.dtype) * rescale_grad
        if clip_gradient > 0:
            g = g.clamp(-clip_gradient, clip_gradient)
        g = g + wd * weight.to(mean.dtype)
        mean.mul_(beta1).add_(g, alpha=1 - beta1)
        var.mul_(beta2).addcmul_(g, g, value=1 - beta2)
        upd = lr * mean / (var.sqrt() + epsilon)
        weight.sub_(upd.to(weight.dtype))
    return weight


@register("mp_adam_update")
def mp_adam_update(weight, grad, mean, var, weight32, lr, beta1=0.9,
                   beta2=0.999, epsilon=1e-8, wd=0.0, rescale_grad=1.0,
                   clip_gradient=-1.0, out=None):
    with torch.no_grad():
        if _maybe_hip("mp_adam_update", weight):
            gpu_impl("mp_adam_update")(weight, grad, mean, var, weight32, lr,
                                       beta1, beta2, epsilon, wd, rescale_grad,
                                       clip_gradient)
            return weight
        g = grad.float() * rescale_grad
        if clip_gradient > 0:
            g = g.clamp(-clip_gradient, clip_gradient)
        g = g + wd * weight32
        mean.mul_(beta1).add_(g, alpha=1 - beta1)
        var.mul_(beta2).addcmul_(g, g, value=1 - beta2)
        weight32.sub_(lr * mean / (var.sqrt() + epsilon))
        weight.copy_(weight32.to(weight.dtype))
    return weight


@register("lamb_update_phase1")
def lamb_update_phase1(weight, grad, mean, var, beta1=0.9, beta2=0.999,
                       epsilon=1e-6, t=1, bias_correction=True, wd=0.0,
                       rescale_grad=1.0, clip_gradient=-1.0):
    with torch.no_grad():
        g = grad.float() * rescale_grad
        if clip_gradient > 0:
            g = g.clamp(-clip_gradient, clip_gradient)
        mean.mul_(beta1).add_(g, alpha=1 - beta1)
        var.mul_(beta2).addcmul_(g, g, value=1 - beta2)
        m, v = mean, var
        if bias_correction:
            m = mean / (1 - beta1 ** t)
            v = var / (1 - beta2 ** t)
        return m / (v.sqrt() + epsilon) + wd * weight.float()


@register("lamb_update_phase2")
def lamb_update_phase2(weight, g_update, r1, r2, lr, lower_bound=-1.0,
                       upper_bound=-1.0):
    with torch.no_grad():
        r1v = r1.item() if isinstance(r1, torch.Tensor) else r1
        r2v = r2.item() if isinstance(r2, torch.Tensor) else r2
        if lower_bound > 0:
            r1v = max(r1v, lower_bound)
        if upper_bound > 0:
            r1v = min(r1v, upper_bound)
        ratio = 1.0 if (r1v == 0.0 or r2v == 0.0) else r1v / r2v
        weight.sub_((lr * ratio * g_update).to(weight.dtype))
    return weight


# ---------------- multi-tensor (fused) variants ----------------

@register("multi_sgd_mom_update")
def multi_sgd_mom_update(weights, grads, moms, lrs, wds, momentum=0.0,
                         rescale_grad=1.0, clip_gradient=-1.0):
    if weights and _maybe_hip("multi_sgd_mom_update", weights[0]):
        gpu_impl("multi_sgd_mom_update")(weights, grads, moms, lrs, wds,
                                         momentum, rescale_grad, clip_gradient)
        return
    for w, g, m, lr, wd in zip(weights, grads, moms, lrs, wds):
        sgd_mom_update(w, g, m, lr, momentum=momentum, wd=wd,
                       rescale_grad=rescale_grad, clip_gradient=clip_gradient)


@register("multi_mp_sgd_mom_update")
def multi_mp_sgd_mom_update(weights, grads, moms, weights32, lrs, wds,
                            momentum=0.0, rescale_grad=1.0,
                            clip_gradient=-1.0):
    if weights and _maybe_hip("multi_mp_sgd_mom_update", weights[0]):
        gpu_impl("multi_mp_sgd_mom_update")(weights, grads, moms, weights32,
                                            lrs, wds, momentum, rescale_grad,
                                            clip_gradient)
        return
    for w, g, m, w32, lr, wd in zip(weights, grads, moms, weights32, lrs, wds):
        mp_sgd_mom_update(w, g, m, w32, lr, momentum=momentum, wd=wd,
                          rescale_grad=rescale_grad,
                          clip_gradient=clip_gradient)


@register("multi_adam_update")
def multi_adam_update(weights, grads, means, vars_, lrs, wds, beta1=0.9,
                      beta2=0.999, epsilon=1e-8, rescale_grad=1.0,
                      clip_gradient=-1.0):
    if weights and _maybe_hip("multi_adam_update", weights[0]):
        gpu_impl("multi_adam_update")(weights, grads, means, vars_, lrs, wds,
                                      beta1, beta2, epsilon, rescale_grad,
                                      clip_gradient)
        return
    for w, g, m, v, lr, wd in zip(weights, grads, means, vars_, lrs, wds):
        adam_update(w, g, m, v, lr, beta1=beta1, beta2=beta2, epsilon=epsilon,
                    wd=wd, rescale_grad=rescale_grad,
                    clip_gradient=clip_gradient)


@register("multi_sum_sq")
def multi_sum_sq(arrays):
    """Per-tensor sum of squares (for LAMB / global-norm clip)."""
    if arrays and _maybe_hip("multi_sum_sq", arrays[0]):
        return gpu_impl("multi_sum_sq")(arrays)
    return [a.float().pow(2).sum() for a in arrays]


@register("adamw_update")
def adamw_update(weight, grad, mean, var, rescale_grad, lr, beta1=0.9,
                 beta2=0.999, epsilon=1e-8, wd=0.0, eta=1.0,
                 clip_gradient=-1.0, out=None):
    """AdamW with decoupled weight decay (reference:
    src/operator/contrib/adamw.cc — note rescale_grad is a TENSOR there
    so a dynamic loss scale can divide on device; scalar also accepted).
    update = eta * (lr * m/(sqrt(v)+eps) + wd * w)."""
    with torch.no_grad():
        rs = rescale_grad if not torch.is_tensor(rescale_grad) \
            else rescale_grad.item()
        g = grad.to(mean.dtype) * rs
        if clip_gradient > 0:
            g = g.clamp(-clip_gradient, clip_gradient)
        mean.mul_(beta1).add_(g, alpha=1 - beta1)
        var.mul_(beta2).addcmul_(g, g, value=1 - beta2)
        w32 = weight.to(mean.dtype)
        upd = lr * mean / (var.sqrt() + epsilon) + wd * w32
        weight.copy_((w32 - eta * upd).to(weight.dtype))
    return weight


@register("mp_adamw_update")
def mp_adamw_update(weight, grad, mean, var, weight32, rescale_grad, lr,
                    beta1=0.9, beta2=0.999, epsilon=1e-8, wd=0.0, eta=1.0,
                    clip_gradient=-1.0, out=None):
    """Multi-precision AdamW: fp32 master weights (reference contrib)."""
    with torch.no_grad():
        rs = rescale_grad if not torch.is_tensor(rescale_grad) \
            else rescale_grad.item()
        g = grad.float() * rs
        if clip_gradient > 0:
            g = g.clamp(-clip_gradient, clip_gradient)
        mean.mul_(beta1).add_(g, alpha=1 - beta1)
        var.mul_(beta2).addcmul_(g, g, value=1 - beta2)
        upd = lr * mean / (var.sqrt() + epsilon) + wd * weight32
        weight32.sub_(eta * upd)
        weight.copy_(weight32.to(weight.dtype))
    return weight
