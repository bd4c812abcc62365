"""Test utilities (parity with python/mxnet/test_utils.py:
assert_almost_equal, check_numeric_gradient (finite differences vs
autograd), check_consistency (cpu vs gpu — the oracle pattern for every HIP
kernel), default contexts, with_seed lives in tests/conftest.py)."""
from __future__ import annotations

import numpy as np

from .context import Context, cpu, gpu, current_context
from .ndarray.ndarray import NDArray, array
from . import autograd as ag

__all__ = ["default_context", "set_default_context", "assert_almost_equal",
           "almost_equal", "check_numeric_gradient", "check_consistency",
           "rand_ndarray", "same"]

_default_ctx = None


def default_context():
    return _default_ctx or current_context()


def set_default_context(ctx):
    global _default_ctx
    _default_ctx = ctx


def same(a, b):
    return np.array_equal(a, b)


def almost_equal(a, b, rtol=1e-5, atol=1e-20):
    return np.allclose(a, b, rtol=rtol, atol=atol)


def _to_np(x):
    if isinstance(x, NDArray):
        return x.asnumpy()
    return np.asarray(x)


def assert_almost_equal(a, b, rtol=1e-5, atol=1e-20, names=("a", "b")):
    a, b = _to_np(a), _to_np(b)
    if not np.allclose(a, b, rtol=rtol, atol=atol):
        err = np.abs(a - b)
        rel = err / (np.abs(b) + atol)
        raise AssertionError(
            f"{names[0]} and {names[1]} differ: max abs err "
            f"{err.max():.3e}, max rel err {rel.max():.3e} "
            f"(rtol={rtol}, atol={atol})")


def rand_ndarray(shape, dtype=np.float32, ctx=None, scale=1.0):
    return array(np.random.uniform(-scale, scale, size=shape).astype(dtype),
                 ctx=ctx)


def check_numeric_gradient(fn, inputs, eps=1e-3, rtol=1e-2, atol=1e-4):
    """Finite differences vs autograd (reference test_utils pattern).

    fn: callable taking NDArrays, returning a scalar-reducible NDArray.
    inputs: list of NDArrays (fp32/fp64 recommended).
    """
    for x in inputs:
        x.attach_grad()
    with ag.record():
        out = fn(*inputs)
        loss = out.sum() if out.size > 1 else out
    loss.backward()
    analytic = [x.grad.asnumpy().copy() for x in inputs]

    for xi, x in enumerate(inputs):
        xn = x.asnumpy()
        numeric = np.zeros_like(xn, dtype=np.float64)
        flat = xn.reshape(-1)
        num_flat = numeric.reshape(-1)
        for i in range(flat.size):
            orig = flat[i]
            flat[i] = orig + eps
            x[...] = xn
            with ag.pause():
                fp = fn(*inputs).sum().asscalar()
            flat[i] = orig - eps
            x[...] = xn
            with ag.pause():
                fm = fn(*inputs).sum().asscalar()
            flat[i] = orig
            x[...] = xn
            num_flat[i] = (fp - fm) / (2 * eps)
        assert_almost_equal(analytic[xi], numeric, rtol=rtol, atol=atol,
                            names=(f"autograd_grad[{xi}]",
                                   f"numeric_grad[{xi}]"))


def check_consistency(fn, inputs_np, ctx_list=None, rtol=None, atol=None,
                      grad=True, dtypes=None):
    """Run fn on each ctx/dtype; compare outputs & grads against the first
    (cpu/fp32) result — the reference's gpu-vs-cpu oracle."""
    import torch
    if ctx_list is None:
        ctx_list = [cpu()] + ([gpu(0)] if torch.cuda.is_available() else [])
    dtypes = dtypes or ["float32"]
    base_out = base_grads = None
    for ctx in ctx_list:
        for dt in dtypes:
            tol_r = rtol if rtol is not None else \
                (1e-4 if dt == "float32" else 3e-2)
            tol_a = atol if atol is not None else \
                (1e-5 if dt == "float32" else 3e-2)
            xs = [array(a, ctx=ctx, dtype=dt) for a in inputs_np]
            if grad:
                for x in xs:
                    x.attach_grad()
                with ag.record():
                    out = fn(*xs)
                out.backward(array(np.ones(out.shape, np.float32), ctx=ctx,
                                   dtype=dt))
                grads = [x.grad.asnumpy() for x in xs]
            else:
                with ag.pause():
                    out = fn(*xs)
                grads = []
            o = out.asnumpy()
            if base_out is None:
                base_out, base_grads = o, grads
            else:
                assert_almost_equal(o, base_out, rtol=tol_r, atol=tol_a,
                                    names=(f"{ctx}/{dt}", "baseline"))
                for g, bg in zip(grads, base_grads):
                    assert_almost_equal(g, bg, rtol=tol_r, atol=tol_a,
                                        names=(f"grad {ctx}/{dt}",
                                               "baseline"))
    return base_out


def retry(n):
    """Retry decorator for stochastic tests (reference:
    tests/python/unittest/common.py — retry)."""
    def deco(fn):
        import functools

        @functools.wraps(fn)
        def wrapped(*args, **kwargs):
            last = None
            for _ in range(n):
                try:
                    return fn(*args, **kwargs)
                except AssertionError as e:
                    last = e
            raise last
        return wrapped
    return deco
