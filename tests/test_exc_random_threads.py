"""Exception propagation, random-moment, and thread-local-scope tests
(reference parity: tests/python/unittest/test_exc_handling.py,
test_random.py, test_thread_local.py)."""
import threading

import numpy as np
import pytest

import anand_mxnet_amd as mx
from anand_mxnet_amd import nd
from anand_mxnet_amd.base import MXNetError


def test_op_error_propagates():
    a = nd.ones((2, 3))
    b = nd.ones((4, 5))
    with pytest.raises((MXNetError, RuntimeError)):
        nd.dot(a, b).asnumpy()  # shape mismatch surfaces at/at-latest sync


def test_executor_error_propagates():
    data = mx.sym.var("data")
    out = mx.sym.FullyConnected(data=data, weight=mx.sym.var("w"),
                                bias=mx.sym.var("b"), num_hidden=4)
    ex = out.simple_bind(mx.cpu(), data=(2, 8))
    with pytest.raises((MXNetError, RuntimeError, KeyError)):
        # forward without binding a required input
        bad = mx.sym.var("missing")
        (data + bad).eval({"data": nd.ones((2, 8))})
    del ex


def test_random_moments():
    mx.random.seed(42)
    u = nd.random.uniform(0, 1, shape=(200000,))
    un = u.asnumpy()
    assert abs(un.mean() - 0.5) < 0.01
    assert abs(un.var() - 1.0 / 12) < 0.005
    n = nd.random.normal(2.0, 3.0, shape=(200000,))
    nn_ = n.asnumpy()
    assert abs(nn_.mean() - 2.0) < 0.05
    assert abs(nn_.std() - 3.0) < 0.05
    p = nd.random_poisson(lam=4.0, shape=(100000,))
    pn = p.asnumpy()
    assert abs(pn.mean() - 4.0) < 0.1
    assert abs(pn.var() - 4.0) < 0.3
    g = nd.random_gamma(alpha=9.0, beta=0.5, shape=(100000,))
    gn = g.asnumpy()
    assert abs(gn.mean() - 4.5) < 0.1


def test_random_seed_reproducible():
    mx.random.seed(7)
    a = nd.random.uniform(shape=(16,)).asnumpy()
    mx.random.seed(7)
    b = nd.random.uniform(shape=(16,)).asnumpy()
    np.testing.assert_array_equal(a, b)


def test_scopes_are_thread_local():
    from anand_mxnet_amd import attribute, name

    results = {}

    def worker():
        # this thread sees a fresh scope, not main's
        s = mx.sym.var("a") + mx.sym.var("b")
        results["thread_attr"] = s.attrs.get("ctx_group")
        with name.Prefix("w_"):
            s2 = mx.sym.var("c") + mx.sym.var("d")
            results["thread_name"] = s2.name

    with attribute.AttrScope(ctx_group="dev9"):
        with name.Prefix("m_"):
            t = threading.Thread(target=worker)
            t.start()
            t.join()
            s_main = mx.sym.var("e") + mx.sym.var("f")
    assert s_main.attrs.get("ctx_group") == "dev9"
    assert s_main.name.startswith("m_")
    assert results["thread_attr"] is None
    assert results["thread_name"].startswith("w_")
    assert not results["thread_name"].startswith("m_w_")
