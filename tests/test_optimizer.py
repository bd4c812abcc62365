"""Optimizer tests vs slow NumPy references (reference pattern:
tests/python/unittest/test_optimizer.py)."""
import numpy as np
import pytest

import anand_mxnet_amd as mx
from anand_mxnet_amd import nd, optimizer as opt
from anand_mxnet_amd.test_utils import assert_almost_equal


def _run_steps(optimizer, w0, grads):
    w = nd.array(w0.copy())
    state = optimizer.create_state_multi_precision(0, w)
    for g in grads:
        optimizer.update_multi_precision(0, w, nd.array(g), state)
    return w.asnumpy()


def test_sgd_vs_numpy():
    np.random.seed(0)
    w0 = np.random.randn(10).astype(np.float32)
    grads = [np.random.randn(10).astype(np.float32) for _ in range(5)]
    lr, mom, wd = 0.1, 0.9, 0.01
    got = _run_steps(opt.SGD(learning_rate=lr, momentum=mom, wd=wd),
                     w0, grads)
    w, m = w0.copy(), np.zeros_like(w0)
    for g in grads:
        gg = g + wd * w
        m = mom * m - lr * gg
        w = w + m
    assert_almost_equal(got, w, rtol=1e-5, atol=1e-6)


def test_sgd_multi_precision_bf16():
    np.random.seed(1)
    w0 = np.random.randn(32).astype(np.float32)
    grads = [np.random.randn(32).astype(np.float32) for _ in range(5)]
    o = opt.SGD(learning_rate=0.1, momentum=0.9, multi_precision=True)
    w = nd.array(w0).astype("bfloat16")
    state = o.create_state_multi_precision(0, w)
    assert isinstance(state, tuple)  # (fp32 master, mom)
    for g in grads:
        o.update_multi_precision(0, w, nd.array(g).astype("bfloat16"), state)
    # fp32 reference with bf16 grad quantization
    import torch
    wr = torch.from_numpy(w0).bfloat16().float()  # master from bf16 w
    m = torch.zeros(32)
    for g in grads:
        gq = torch.from_numpy(g).bfloat16().float()
        m = 0.9 * m - 0.1 * gq
        wr = wr + m
    assert_almost_equal(state[0].numpy(), wr.numpy(), rtol=1e-4, atol=1e-4)


def test_adam_vs_numpy():
    np.random.seed(2)
    w0 = np.random.randn(16).astype(np.float32)
    grads = [np.random.randn(16).astype(np.float32) for _ in range(4)]
    lr, b1, b2, eps = 0.01, 0.9, 0.999, 1e-8
    got = _run_steps(opt.Adam(learning_rate=lr, beta1=b1, beta2=b2,
                              epsilon=eps), w0, grads)
    w = w0.copy()
    m = np.zeros_like(w0)
    v = np.zeros_like(w0)
    for t, g in enumerate(grads, 1):
        m = b1 * m + (1 - b1) * g
        v = b2 * v + (1 - b2) * g * g
        lr_t = lr * np.sqrt(1 - b2 ** t) / (1 - b1 ** t)
        w -= lr_t * m / (np.sqrt(v) + eps)
    assert_almost_equal(got, w, rtol=1e-4, atol=1e-5)


def test_rmsprop_adagrad_run():
    for name in ("rmsprop", "adagrad", "ftrl", "signum", "nag", "adamw",
                 "lamb"):
        o = opt.create(name)
        w = nd.ones((4,))
        state = o.create_state_multi_precision(0, w)
        o.update_multi_precision(0, w, nd.ones((4,)) * 0.1, state)
        assert np.all(np.isfinite(w.asnumpy()))


def test_lr_scheduler():
    from anand_mxnet_amd.lr_scheduler import (FactorScheduler,
                                              MultiFactorScheduler,
                                              PolyScheduler, CosineScheduler)
    s = FactorScheduler(step=10, factor=0.5, base_lr=1.0)
    assert s(5) == 1.0
    assert s(15) == 0.5
    s2 = MultiFactorScheduler(step=[5, 10], factor=0.1, base_lr=1.0)
    assert s2(1) == 1.0
    assert abs(s2(7) - 0.1) < 1e-9
    assert abs(s2(11) - 0.01) < 1e-9
    s3 = PolyScheduler(max_update=100, base_lr=1.0, pwr=1)
    assert abs(s3(50) - 0.5) < 1e-6
    s4 = CosineScheduler(max_update=100, base_lr=1.0)
    assert abs(s4(50) - 0.5) < 1e-6
    warm = FactorScheduler(step=1000, base_lr=1.0, warmup_steps=10)
    assert warm(5) == 0.5


def test_optimizer_with_scheduler_in_trainer():
    from anand_mxnet_amd import gluon, autograd
    from anand_mxnet_amd.lr_scheduler import FactorScheduler
    net = gluon.nn.Dense(2, in_units=2)
    net.initialize()
    tr = gluon.Trainer(net.collect_params(), "sgd",
                       {"learning_rate": 1.0,
                        "lr_scheduler": FactorScheduler(step=2, factor=0.1)})
    x = nd.ones((1, 2))
    for i in range(4):
        with autograd.record():
            loss = net(x).sum()
        loss.backward()
        tr.step(1)
    assert tr.learning_rate < 1.0


def test_updater_states_roundtrip():
    o = opt.Adam()
    upd = opt.get_updater(o)
    w = nd.ones((4,))
    upd(0, nd.ones((4,)), w)
    blob = upd.get_states()
    upd2 = opt.get_updater(opt.Adam())
    upd2.set_states(blob)
    assert 0 in upd2.states


def test_wd_mult_lr_mult():
    o = opt.SGD(learning_rate=1.0)
    o.set_lr_mult({0: 0.5})
    o.set_wd_mult({0: 0.0})
    assert o._get_lr(0) == 0.5
    assert o._get_wd(0) == 0.0


def test_lamb_vs_numpy():
    """LAMB update vs a slow numpy reference (reference: optimizer.py
    LAMB / lamb_update_phase1+2)."""
    from anand_mxnet_amd import optimizer as opt
    rs = np.random.RandomState(0)
    w0 = rs.normal(size=(6,)).astype("float32")
    g0 = rs.normal(size=(6,)).astype("float32")
    lamb = opt.create("lamb", learning_rate=0.01, beta1=0.9, beta2=0.999,
                      epsilon=1e-6, wd=0.01)
    w = nd.array(w0.copy())
    grad = nd.array(g0.copy())
    state = lamb.create_state(0, w)
    lamb.update(0, w, grad, state)
    # numpy reference
    m = (1 - 0.9) * g0
    v = (1 - 0.999) * g0 * g0
    mh = m / (1 - 0.9)
    vh = v / (1 - 0.999)
    upd = mh / (np.sqrt(vh) + 1e-6) + 0.01 * w0
    r1 = np.linalg.norm(w0)
    r2 = np.linalg.norm(upd)
    trust = r1 / r2 if (r1 > 0 and r2 > 0) else 1.0
    ref = w0 - 0.01 * trust * upd
    np.testing.assert_allclose(w.asnumpy(), ref, rtol=1e-4, atol=1e-5)


def test_ftrl_signum_nag_run():
    from anand_mxnet_amd import optimizer as opt
    for name, kw in [("ftrl", {"learning_rate": 0.1}),
                     ("signum", {"learning_rate": 0.01}),
                     ("nag", {"learning_rate": 0.1, "momentum": 0.9})]:
        o = opt.create(name, **kw)
        w = nd.array(np.ones(4, dtype="float32"))
        g = nd.array(np.full(4, 0.5, dtype="float32"))
        st = o.create_state(0, w)
        before = w.asnumpy().copy()
        o.update(0, w, g, st)
        o.update(0, w, g, st)
        assert not np.allclose(w.asnumpy(), before), name


def test_adamw_decoupled_wd():
    """AdamW: wd acts on the weight directly, not through the adaptive
    scaling (reference contrib adamw)."""
    from anand_mxnet_amd import optimizer as opt
    o = opt.create("adamw", learning_rate=0.1, wd=0.5)
    w = nd.array(np.array([1.0], dtype="float32"))
    g = nd.array(np.array([0.0], dtype="float32"))
    st = o.create_state(0, w)
    o.update(0, w, g, st)
    # zero grad: only decay moves the weight: w -= lr * wd * w
    np.testing.assert_allclose(w.asnumpy(), [1.0 - 0.1 * 0.5], rtol=1e-5)


def test_trainer_save_load_states(tmp_path):
    """Trainer.save_states/load_states round-trips optimizer state
    (reference trainer.py: updater states pickled with the optimizer);
    resumed training matches uninterrupted training bit-for-bit."""
    import numpy as np
    import anand_mxnet_amd as mx
    from anand_mxnet_amd import gluon
    from anand_mxnet_amd.gluon import nn

    def make():
        net = nn.Dense(4, in_units=6)
        net.initialize(mx.init.Constant(0.05))
        tr = gluon.Trainer(net.collect_params(), "adam",
                           {"learning_rate": 1e-2})
        return net, tr

    rs = np.random.RandomState(0)
    xs = [mx.nd.array(rs.rand(3, 6).astype("float32")) for _ in range(6)]

    def step(net, tr, x):
        from anand_mxnet_amd import autograd as ag
        with ag.record():
            loss = (net(x) ** 2).mean()
        loss.backward()
        tr.step(1)

    # run A: 6 uninterrupted steps
    netA, trA = make()
    for x in xs:
        step(netA, trA, x)

    # run B: 3 steps, save, rebuild, load, 3 more
    netB, trB = make()
    for x in xs[:3]:
        step(netB, trB, x)
    pfx = str(tmp_path / "t")
    netB.save_parameters(pfx + ".params")
    trB.save_states(pfx + ".states")
    netC, trC = make()
    netC.load_parameters(pfx + ".params")
    trC.load_states(pfx + ".states")
    for x in xs[3:]:
        step(netC, trC, x)

    wA = netA.weight.data().asnumpy()
    wC = netC.weight.data().asnumpy()
    assert np.allclose(wA, wC, atol=1e-6), np.abs(wA - wC).max()


def test_trainer_lr_scheduler_progression():
    """Trainer honors lr_scheduler across steps (reference
    lr_scheduler.py FactorScheduler) and learning_rate reflects it."""
    import numpy as np
    import anand_mxnet_amd as mx
    from anand_mxnet_amd import gluon, lr_scheduler
    from anand_mxnet_amd.gluon import nn
    from anand_mxnet_amd import autograd as ag

    sched = lr_scheduler.FactorScheduler(step=2, factor=0.5,
                                         base_lr=0.4)
    net = nn.Dense(2, in_units=3)
    net.initialize(mx.init.Constant(0.1))
    tr = gluon.Trainer(net.collect_params(), "sgd",
                       {"learning_rate": 0.4, "lr_scheduler": sched})
    x = mx.nd.ones((2, 3))
    lrs = []
    for _ in range(5):
        with ag.record():
            loss = (net(x) ** 2).mean()
        loss.backward()
        tr.step(1)
        lrs.append(tr.learning_rate)
    # factor 0.5 every 2 updates: 0.4, 0.4, 0.2, 0.2, 0.1
    assert np.allclose(lrs, [0.4, 0.4, 0.2, 0.2, 0.1]), lrs
