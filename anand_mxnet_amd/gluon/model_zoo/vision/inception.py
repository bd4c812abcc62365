"""Inception V3 (parity with gluon/model_zoo/vision/inception.py)."""
from ... import nn
from ...block import HybridBlock
from .... import ndarray as F_nd

__all__ = ["Inception3", "inception_v3"]


def _make_basic_conv(channels, **kwargs):
    out = nn.HybridSequential(prefix="")
    out.add(nn.Conv2D(channels, use_bias=False, **kwargs))
    out.add(nn.BatchNorm(epsilon=0.001))
    out.add(nn.Activation("relu"))
    return out


class _Branching(HybridBlock):
    """Concat of parallel branches along channels."""

    def __init__(self, branches, **kwargs):
        super().__init__(**kwargs)
        self._branches = []
        for i, b in enumerate(branches):
            self.register_child(b, f"b{i}")
            self._branches.append(b)

    def forward(self, x):
        return F_nd.concat(*[b(x) for b in self._branches], dim=1)

    hybrid_forward = None


def _make_branch(use_pool, *conv_settings):
    out = nn.HybridSequential(prefix="")
    if use_pool == "avg":
        out.add(nn.AvgPool2D(pool_size=3, strides=1, padding=1))
    elif use_pool == "max":
        out.add(nn.MaxPool2D(pool_size=3, strides=2))
    for setting in conv_settings:
        kwargs = {}
        channels, kernel, strides, padding = setting
        if kernel is not None:
            kwargs["kernel_size"] = kernel
        if strides is not None:
            kwargs["strides"] = strides
        if padding is not None:
            kwargs["padding"] = padding
        out.add(_make_basic_conv(channels, **kwargs))
    return out


def _make_A(pool_features, prefix):
    return _Branching([
        _make_branch(None, (64, 1, None, None)),
        _make_branch(None, (48, 1, None, None), (64, 5, None, 2)),
        _make_branch(None, (64, 1, None, None), (96, 3, None, 1),
                     (96, 3, None, 1)),
        _make_branch("avg", (pool_features, 1, None, None)),
    ], prefix=prefix)


def _make_B(prefix):
    return _Branching([
        _make_branch(None, (384, 3, 2, None)),
        _make_branch(None, (64, 1, None, None), (96, 3, None, 1),
                     (96, 3, 2, None)),
        _make_branch("max"),
    ], prefix=prefix)


def _make_C(channels_7x7, prefix):
    return _Branching([
        _make_branch(None, (192, 1, None, None)),
        _make_branch(None, (channels_7x7, 1, None, None),
                     (channels_7x7, (1, 7), None, (0, 3)),
                     (192, (7, 1), None, (3, 0))),
        _make_branch(None, (channels_7x7, 1, None, None),
                     (channels_7x7, (7, 1), None, (3, 0)),
                     (channels_7x7, (1, 7), None, (0, 3)),
                     (channels_7x7, (7, 1), None, (3, 0)),
                     (192, (1, 7), None, (0, 3))),
        _make_branch("avg", (192, 1, None, None)),
    ], prefix=prefix)


def _make_D(prefix):
    return _Branching([
        _make_branch(None, (192, 1, None, None), (320, 3, 2, None)),
        _make_branch(None, (192, 1, None, None), (192, (1, 7), None, (0, 3)),
                     (192, (7, 1), None, (3, 0)), (192, 3, 2, None)),
        _make_branch("max"),
    ], prefix=prefix)


class _SplitConcat(HybridBlock):
    """x -> base -> concat(sub1(x'), sub2(x')) (inception E inner split)."""

    def __init__(self, base, subs, **kwargs):
        super().__init__(**kwargs)
        self.base = base
        self._subs = []
        for i, sb in enumerate(subs):
            self.register_child(sb, f"s{i}")
            self._subs.append(sb)

    def forward(self, x):
        h = self.base(x) if self.base is not None else x
        return F_nd.concat(*[sb(h) for sb in self._subs], dim=1)

    hybrid_forward = None


def _make_E(prefix):
    return _Branching([
        _make_branch(None, (320, 1, None, None)),
        _SplitConcat(_make_basic_conv(384, kernel_size=1),
                     [_make_basic_conv(384, kernel_size=(1, 3),
                                       padding=(0, 1)),
                      _make_basic_conv(384, kernel_size=(3, 1),
                                       padding=(1, 0))]),
        _SplitConcat(
            nn.HybridSequential(),
            [_make_basic_conv(384, kernel_size=(1, 3), padding=(0, 1)),
             _make_basic_conv(384, kernel_size=(3, 1), padding=(1, 0))]),
        _make_branch("avg", (192, 1, None, None)),
    ], prefix=prefix)


class Inception3(HybridBlock):
    def __init__(self, classes=1000, **kwargs):
        super().__init__(**kwargs)
        with self.name_scope():
            self.features = nn.HybridSequential(prefix="")
            self.features.add(_make_basic_conv(32, kernel_size=3, strides=2))
            self.features.add(_make_basic_conv(32, kernel_size=3))
            self.features.add(_make_basic_conv(64, kernel_size=3, padding=1))
            self.features.add(nn.MaxPool2D(pool_size=3, strides=2))
            self.features.add(_make_basic_conv(80, kernel_size=1))
            self.features.add(_make_basic_conv(192, kernel_size=3))
            self.features.add(nn.MaxPool2D(pool_size=3, strides=2))
            self.features.add(_make_A(32, "A1_"))
            self.features.add(_make_A(64, "A2_"))
            self.features.add(_make_A(64, "A3_"))
            self.features.add(_make_B("B_"))
            self.features.add(_make_C(128, "C1_"))
            self.features.add(_make_C(160, "C2_"))
            self.features.add(_make_C(160, "C3_"))
            self.features.add(_make_C(192, "C4_"))
            self.features.add(_make_D("D_"))
            self.features.add(_make_E("E1_"))
            self.features.add(_make_E("E2_"))
            self.features.add(nn.AvgPool2D(pool_size=8))
            self.features.add(nn.Dropout(0.5))
            self.output = nn.Dense(classes)

    def forward(self, x):
        return self.output(self.features(x))

    hybrid_forward = None


def inception_v3(**kwargs):
    return Inception3(**kwargs)
