"""C predict API (reference c_predict_api.cc parity): the extern "C"
symbols exported from the extension .so, driven through ctypes exactly as
a C program would call them."""
import ctypes
import os

import numpy as np
import pytest


def _clib():
    import anand_mxnet_amd._C as C
    lib = ctypes.CDLL(C.__file__)
    lib.MXAGetLastError.restype = ctypes.c_char_p
    return lib


def _export_model(tmp_path):
    import anand_mxnet_amd as mx
    from anand_mxnet_amd import sym, nd
    from anand_mxnet_amd.module.module import Module
    data = sym.Variable("data")
    net = sym.FullyConnected(data, num_hidden=8, name="fc1")
    net = sym.Activation(net, act_type="relu", name="relu1")
    net = sym.FullyConnected(net, num_hidden=3, name="fc2")
    net = sym.SoftmaxOutput(net, name="softmax")
    mod = Module(net, data_names=("data",), label_names=("softmax_label",))
    mod.bind([("data", (2, 5))], [("softmax_label", (2,))],
             for_training=False)
    mod.init_params(mx.init.Xavier())
    prefix = str(tmp_path / "m")
    mod.save_checkpoint(prefix, 0)
    return prefix, mod


def test_predict_matches_module(tmp_path):
    prefix, mod = _export_model(tmp_path)
    lib = _clib()
    h = ctypes.c_void_p()
    names = (ctypes.c_char_p * 1)(b"data")
    rc = lib.MXAPredCreate(
        (prefix + "-symbol.json").encode(), (prefix + "-0000.params").encode(),
        1, 0, 1, names, ctypes.byref(h))
    assert rc == 0, lib.MXAGetLastError()

    x = np.random.RandomState(0).uniform(-1, 1, (2, 5)).astype(np.float32)
    shape = (ctypes.c_long * 2)(2, 5)
    rc = lib.MXAPredSetInput(h, b"data",
                             x.ctypes.data_as(ctypes.POINTER(ctypes.c_float)),
                             shape, 2)
    assert rc == 0, lib.MXAGetLastError()
    rc = lib.MXAPredForward(h)
    assert rc == 0, lib.MXAGetLastError()

    oshape = (ctypes.c_long * 8)()
    ondim = ctypes.c_uint()
    rc = lib.MXAPredGetOutputShape(h, 0, oshape, ctypes.byref(ondim))
    assert rc == 0, lib.MXAGetLastError()
    dims = tuple(oshape[i] for i in range(ondim.value))
    assert dims == (2, 3)
    out = np.zeros(dims, dtype=np.float32)
    rc = lib.MXAPredGetOutput(
        h, 0, out.ctypes.data_as(ctypes.POINTER(ctypes.c_float)),
        out.size)
    assert rc == 0, lib.MXAGetLastError()
    assert lib.MXAPredFree(h) == 0

    # oracle: the same checkpoint through the Module API
    from anand_mxnet_amd import nd
    from anand_mxnet_amd.io import DataBatch
    from anand_mxnet_amd.module.module import Module as M2
    m2 = M2.load(prefix, 0, data_names=("data",), label_names=None)
    m2.bind([("data", (2, 5))], for_training=False)
    m2.init_from_loaded()
    m2.forward(DataBatch(data=[nd.array(x)]), is_train=False)
    ref = m2.get_outputs()[0].asnumpy()
    np.testing.assert_allclose(out, ref, rtol=1e-5, atol=1e-6)


def test_predict_error_contract(tmp_path):
    lib = _clib()
    h = ctypes.c_void_p()
    names = (ctypes.c_char_p * 1)(b"data")
    rc = lib.MXAPredCreate(b"/nonexistent-symbol.json", b"/nope.params",
                           1, 0, 1, names, ctypes.byref(h))
    assert rc == -1
    assert len(lib.MXAGetLastError()) > 0
