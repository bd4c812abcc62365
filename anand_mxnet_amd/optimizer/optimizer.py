"""Optimizers (parity with python/mxnet/optimizer/optimizer.py: registry,
SGD w/ momentum + multi-precision, Adam, LAMB, NAG, RMSProp, AdaGrad, Ftrl,
Signum, Updater for kvstore-side application).

Update arithmetic is delegated to the optimizer update ops
(ops/optim.py -> fused multi-tensor HIP kernels on GPU)."""
from __future__ import annotations

import math
import pickle

import numpy as np
import torch

from ..base import MXNetError
from ..ndarray.ndarray import NDArray
from ..ops import registry as _reg

__all__ = ["Optimizer", "SGD", "NAG", "Adam", "AdamW", "LAMB", "RMSProp",
           "AdaGrad", "Ftrl", "Signum", "create", "register", "Updater",
           "get_updater"]

_OPT_REGISTRY = {}


def register(cls):
    _OPT_REGISTRY[cls.__name__.lower()] = cls
    return cls


def create(name, **kwargs):
    if isinstance(name, Optimizer):
        return name
    try:
        return _OPT_REGISTRY[name.lower()](**kwargs)
    except KeyError:
        raise MXNetError(f"unknown optimizer {name}")


class Optimizer:
    def __init__(self, rescale_grad=1.0, param_idx2name=None, wd=0.0,
                 clip_gradient=None, learning_rate=0.01, lr_scheduler=None,
                 sym=None, begin_num_update=0, multi_precision=False,
                 param_dict=None, aggregate_num=None):
        self.rescale_grad = rescale_grad
        self.lr = learning_rate
        self.lr_scheduler = lr_scheduler
        if lr_scheduler is not None:
            self.lr_scheduler.base_lr = learning_rate
        self.wd = wd
        self.lr_mult = {}
        self.wd_mult = {}
        self.begin_num_update = begin_num_update
        self.num_update = begin_num_update
        self._index_update_count = {}
        self.clip_gradient = clip_gradient
        self.multi_precision = multi_precision
        self.param_idx2name = param_idx2name or {}
        self.param_dict = param_dict or {}
        self.aggregate_num = aggregate_num or 1
        self.idx2name = self.param_idx2name

    # ------------- lr/wd bookkeeping -------------
    def set_learning_rate(self, lr):
        if self.lr_scheduler is not None:
            raise UserWarning("lr_scheduler is set; use it instead")
        self.lr = lr

    @property
    def learning_rate(self):
        if self.lr_scheduler is not None:
            return self.lr_scheduler(self.num_update)
        return self.lr

    def set_lr_mult(self, args_lr_mult):
        self.lr_mult = dict(args_lr_mult)

    def set_wd_mult(self, args_wd_mult):
        self.wd_mult = dict(args_wd_mult)

    def _update_count(self, index):
        if not isinstance(index, (list, tuple)):
            index = [index]
        for idx in index:
            if idx not in self._index_update_count:
                self._index_update_count[idx] = self.begin_num_update
            self._index_update_count[idx] += 1
            self.num_update = max(self._index_update_count[idx],
                                  self.num_update)

    def _get_lr(self, index):
        lr = self.lr_scheduler(self.num_update) if self.lr_scheduler else self.lr
        name = self.param_idx2name.get(index, index)
        if name in self.param_dict:
            lr *= getattr(self.param_dict[name], "lr_mult", 1.0)
        lr *= self.lr_mult.get(name, self.lr_mult.get(index, 1.0))
        return lr

    def _get_wd(self, index):
        wd = self.wd
        name = self.param_idx2name.get(index, index)
        if name in self.param_dict:
            wd *= getattr(self.param_dict[name], "wd_mult", 1.0)
        wd *= self.wd_mult.get(name, self.wd_mult.get(index, 1.0))
        return wd

    # ------------- state -------------
    def create_state(self, index, weight):
        return None

    def _needs_master(self, weight):
        return self.multi_precision and weight._t.dtype in (
            torch.float16, torch.bfloat16)

    def create_state_multi_precision(self, index, weight):
        if self._needs_master(weight):
            w32 = weight._t.detach().float().clone()
            return (w32, self.create_state(index, NDArray(w32)))
        return self.create_state(index, weight)

    def update(self, index, weight, grad, state):
        raise NotImplementedError

    def update_multi_precision(self, index, weight, grad, state):
        if self._needs_master(weight):
            w32, inner = state
            self._update_mp(index, weight, grad, inner, w32)
        else:
            self.update(index, weight, grad, state)

    def _update_mp(self, index, weight, grad, state, w32):
        # generic fallback: update fp32 master then copy down
        wrapped = NDArray(w32)
        self.update(index, wrapped, NDArray(grad._t.float()), state)
        with torch.no_grad():
            weight._t.detach().copy_(w32.to(weight._t.dtype))

    # ------------- serialization of states -------------
    def set_states(self, states):
        st = pickle.loads(states)
        self._index_update_count = st.get("update_count", {})
        self.num_update = st.get("num_update", self.num_update)

    def get_states(self, dump_optimizer=False):
        return pickle.dumps({"update_count": self._index_update_count,
                             "num_update": self.num_update})

    def __repr__(self):
        return f"{self.__class__.__name__}(lr={self.lr})"


@register
class SGD(Optimizer):
    """SGD with momentum; multi-precision (fp32 master weights for bf16/fp16);
    fused multi-tensor path used by Trainer (reference multi_sgd_mom_update)."""

    def __init__(self, momentum=0.0, lazy_update=True, **kwargs):
        super().__init__(**kwargs)
        self.momentum = momentum
        self.lazy_update = lazy_update

    def create_state(self, index, weight):
        if self.momentum == 0.0:
            return None
        t = weight._t
        return torch.zeros_like(t.detach(), dtype=torch.float32
                                if t.dtype in (torch.float16, torch.bfloat16)
                                else t.dtype)

    def update(self, index, weight, grad, state):
        self._update_count(index)
        lr, wd = self._get_lr(index), self._get_wd(index)
        clip = self.clip_gradient if self.clip_gradient else -1.0
        if state is None:
            _reg.RAW["sgd_update"](weight._t, grad._t, lr, wd,
                                   self.rescale_grad, clip)
        else:
            _reg.RAW["sgd_mom_update"](weight._t, grad._t, state, lr,
                                       self.momentum, wd, self.rescale_grad,
                                       clip)

    def _update_mp(self, index, weight, grad, state, w32):
        self._update_count(index)
        lr, wd = self._get_lr(index), self._get_wd(index)
        clip = self.clip_gradient if self.clip_gradient else -1.0
        mom = state if state is not None else torch.zeros_like(w32)
        _reg.RAW["mp_sgd_mom_update"](weight._t, grad._t, mom, w32, lr,
                                      self.momentum, wd, self.rescale_grad,
                                      clip)


@register
class NAG(SGD):
    def update(self, index, weight, grad, state):
        self._update_count(index)
        lr, wd = self._get_lr(index), self._get_wd(index)
        with torch.no_grad():
            g = grad._t.float() * self.rescale_grad
            if self.clip_gradient:
                g = g.clamp(-self.clip_gradient, self.clip_gradient)
            g = g + wd * weight._t.float()
            if state is not None:
                state.mul_(self.momentum).add_(g)
                g = g + self.momentum * state
            weight._t.detach().sub_((lr * g).to(weight._t.dtype))


@register
class Adam(Optimizer):
    def __init__(self, learning_rate=0.001, beta1=0.9, beta2=0.999,
                 epsilon=1e-8, lazy_update=True, **kwargs):
        super().__init__(learning_rate=learning_rate, **kwargs)
        self.beta1, self.beta2, self.epsilon = beta1, beta2, epsilon

    def create_state(self, index, weight):
        t = weight._t
        dt = torch.float32 if t.dtype in (torch.float16, torch.bfloat16) \
            else t.dtype
        return (torch.zeros_like(t.detach(), dtype=dt),
                torch.zeros_like(t.detach(), dtype=dt))

    def update(self, index, weight, grad, state):
        self._update_count(index)
        tcount = self._index_update_count[index]
        lr = self._get_lr(index) * math.sqrt(1 - self.beta2 ** tcount) \
            / (1 - self.beta1 ** tcount)
        wd = self._get_wd(index)
        clip = self.clip_gradient if self.clip_gradient else -1.0
        mean, var = state
        _reg.RAW["adam_update"](weight._t, grad._t, mean, var, lr, self.beta1,
                                self.beta2, self.epsilon, wd,
                                self.rescale_grad, clip)

    def _update_mp(self, index, weight, grad, state, w32):
        self._update_count(index)
        tcount = self._index_update_count[index]
        lr = self._get_lr(index) * math.sqrt(1 - self.beta2 ** tcount) \
            / (1 - self.beta1 ** tcount)
        wd = self._get_wd(index)
        clip = self.clip_gradient if self.clip_gradient else -1.0
        mean, var = state
        _reg.RAW["mp_adam_update"](weight._t, grad._t, mean, var, w32, lr,
                                   self.beta1, self.beta2, self.epsilon, wd,
                                   self.rescale_grad, clip)


@register
class AdamW(Adam):
    """Adam with decoupled weight decay (reference contrib adamw_update)."""

    def update(self, index, weight, grad, state):
        self._update_count(index)
        tcount = self._index_update_count[index]
        lr = self._get_lr(index)
        corr = math.sqrt(1 - self.beta2 ** tcount) / (1 - self.beta1 ** tcount)
        wd = self._get_wd(index)
        mean, var = state
        with torch.no_grad():
            g = grad._t.to(mean.dtype) * self.rescale_grad
            if self.clip_gradient:
                g = g.clamp(-self.clip_gradient, self.clip_gradient)
            mean.mul_(self.beta1).add_(g, alpha=1 - self.beta1)
            var.mul_(self.beta2).addcmul_(g, g, value=1 - self.beta2)
            upd = lr * (corr * mean / (var.sqrt() + self.epsilon)
                        + wd * weight._t.to(mean.dtype))
            weight._t.detach().sub_(upd.to(weight._t.dtype))


@register
class LAMB(Optimizer):
    """LAMB (reference: lamb_update_phase1/2 in optimizer_op.cu, 1.6)."""

    def __init__(self, learning_rate=0.001, beta1=0.9, beta2=0.999,
                 epsilon=1e-6, lower_bound=None, upper_bound=None,
                 bias_correction=True, **kwargs):
        super().__init__(learning_rate=learning_rate, **kwargs)
        self.beta1, self.beta2, self.epsilon = beta1, beta2, epsilon
        self.lower_bound, self.upper_bound = lower_bound, upper_bound
        self.bias_correction = bias_correction

    def create_state(self, index, weight):
        t = weight._t
        return (torch.zeros_like(t.detach(), dtype=torch.float32),
                torch.zeros_like(t.detach(), dtype=torch.float32))

    def update(self, index, weight, grad, state):
        self._update_count(index)
        tcount = self._index_update_count[index]
        lr, wd = self._get_lr(index), self._get_wd(index)
        clip = self.clip_gradient if self.clip_gradient else -1.0
        mean, var = state
        g = _reg.RAW["lamb_update_phase1"](
            weight._t, grad._t, mean, var, self.beta1, self.beta2,
            self.epsilon, tcount, self.bias_correction, wd,
            self.rescale_grad, clip)
        r1 = weight._t.detach().float().norm()
        r2 = g.norm()
        _reg.RAW["lamb_update_phase2"](
            weight._t, g, r1, r2, lr,
            self.lower_bound if self.lower_bound else -1.0,
            self.upper_bound if self.upper_bound else -1.0)


@register
class RMSProp(Optimizer):
    def __init__(self, learning_rate=0.001, gamma1=0.9, gamma2=0.9,
                 epsilon=1e-8, centered=False, **kwargs):
        super().__init__(learning_rate=learning_rate, **kwargs)
        self.gamma1, self.gamma2 = gamma1, gamma2
        self.epsilon, self.centered = epsilon, centered

    def create_state(self, index, weight):
        t = weight._t.detach()
        if self.centered:
            return (torch.zeros_like(t, dtype=torch.float32),
                    torch.zeros_like(t, dtype=torch.float32),
                    torch.zeros_like(t, dtype=torch.float32))
        return torch.zeros_like(t, dtype=torch.float32)

    def update(self, index, weight, grad, state):
        self._update_count(index)
        lr, wd = self._get_lr(index), self._get_wd(index)
        with torch.no_grad():
            g = grad._t.float() * self.rescale_grad
            if self.clip_gradient:
                g = g.clamp(-self.clip_gradient, self.clip_gradient)
            g = g + wd * weight._t.float()
            if self.centered:
                n, gm, delta = state
                n.mul_(self.gamma1).addcmul_(g, g, value=1 - self.gamma1)
                gm.mul_(self.gamma1).add_(g, alpha=1 - self.gamma1)
                delta.mul_(self.gamma2).sub_(
                    lr * g / torch.sqrt(n - gm * gm + self.epsilon))
                weight._t.detach().add_(delta.to(weight._t.dtype))
            else:
                n = state
                n.mul_(self.gamma1).addcmul_(g, g, value=1 - self.gamma1)
                weight._t.detach().sub_(
                    (lr * g / torch.sqrt(n + self.epsilon)).to(weight._t.dtype))


@register
class AdaGrad(Optimizer):
    def __init__(self, learning_rate=0.01, eps=1e-7, **kwargs):
        super().__init__(learning_rate=learning_rate, **kwargs)
        self.float_stable_eps = eps

    def create_state(self, index, weight):
        return torch.zeros_like(weight._t.detach(), dtype=torch.float32)

    def update(self, index, weight, grad, state):
        self._update_count(index)
        lr, wd = self._get_lr(index), self._get_wd(index)
        with torch.no_grad():
            g = grad._t.float() * self.rescale_grad
            if self.clip_gradient:
                g = g.clamp(-self.clip_gradient, self.clip_gradient)
            g = g + wd * weight._t.float()
            state.addcmul_(g, g, value=1.0)
            weight._t.detach().sub_(
                (lr * g / (state.sqrt() + self.float_stable_eps)).to(
                    weight._t.dtype))


@register
class Ftrl(Optimizer):
    def __init__(self, lamda1=0.01, learning_rate=0.1, beta=1, **kwargs):
        super().__init__(learning_rate=learning_rate, **kwargs)
        self.lamda1, self.beta = lamda1, beta

    def create_state(self, index, weight):
        t = weight._t.detach()
        return (torch.zeros_like(t, dtype=torch.float32),
                torch.zeros_like(t, dtype=torch.float32))

    def update(self, index, weight, grad, state):
        self._update_count(index)
        lr, wd = self._get_lr(index), self._get_wd(index)
        z, n = state
        with torch.no_grad():
            g = grad._t.float() * self.rescale_grad
            if self.clip_gradient:
                g = g.clamp(-self.clip_gradient, self.clip_gradient)
            w = weight._t.float()
            sigma = (torch.sqrt(n + g * g) - torch.sqrt(n)) / lr
            z.add_(g - sigma * w)
            n.add_(g * g)
            new_w = torch.where(
                torch.abs(z) <= self.lamda1, torch.zeros_like(w),
                (torch.sign(z) * self.lamda1 - z)
                / ((self.beta + torch.sqrt(n)) / lr + wd))
            weight._t.detach().copy_(new_w.to(weight._t.dtype))


@register
class Signum(Optimizer):
    def __init__(self, learning_rate=0.01, momentum=0.9, wd_lh=0.0, **kwargs):
        super().__init__(learning_rate=learning_rate, **kwargs)
        self.momentum = momentum
        self.wd_lh = wd_lh

    def create_state(self, index, weight):
        if self.momentum == 0.0:
            return None
        return torch.zeros_like(weight._t.detach(), dtype=torch.float32)

    def update(self, index, weight, grad, state):
        self._update_count(index)
        lr, wd = self._get_lr(index), self._get_wd(index)
        with torch.no_grad():
            g = grad._t.float() * self.rescale_grad
            if self.clip_gradient:
                g = g.clamp(-self.clip_gradient, self.clip_gradient)
            g = g + wd * weight._t.float()
            if state is not None:
                state.mul_(self.momentum).sub_(g, alpha=1 - self.momentum)
                upd = lr * torch.sign(-state)
            else:
                upd = lr * torch.sign(g)
            weight._t.detach().sub_(upd.to(weight._t.dtype))


class Updater:
    """kvstore-side updater (parity with Optimizer.Updater / get_updater)."""

    def __init__(self, optimizer: Optimizer):
        self.optimizer = optimizer
        self.states = {}
        self.states_synced = {}

    def __call__(self, index, grad, weight):
        if index not in self.states:
            self.states[index] = \
                self.optimizer.create_state_multi_precision(index, weight)
        self.optimizer.update_multi_precision(index, weight, grad,
                                              self.states[index])

    def get_states(self, dump_optimizer=False):
        return pickle.dumps((
            {k: _state_to_numpy(v) for k, v in self.states.items()},
            self.optimizer.get_states() if dump_optimizer else None))

    def set_states(self, states):
        st, opt = pickle.loads(states)
        self.states = {k: _state_from_numpy(v) for k, v in st.items()}
        if opt is not None:
            self.optimizer.set_states(opt)


def _state_to_numpy(s):
    if s is None:
        return None
    if isinstance(s, torch.Tensor):
        return s.detach().cpu().float().numpy() if s.dtype == torch.bfloat16 \
            else s.detach().cpu().numpy()
    if isinstance(s, tuple):
        return tuple(_state_to_numpy(x) for x in s)
    return s


def _state_from_numpy(s):
    if s is None:
        return None
    if isinstance(s, np.ndarray):
        return torch.from_numpy(s.copy())
    if isinstance(s, tuple):
        return tuple(_state_from_numpy(x) for x in s)
    return s


def get_updater(optimizer):
    return Updater(optimizer)
