"""DenseNet (parity with gluon/model_zoo/vision/densenet.py)."""
from ... import nn
from ...block import HybridBlock
from .... import ndarray as F_nd

__all__ = ["DenseNet", "densenet121", "densenet161", "densenet169",
           "densenet201"]


class _DenseLayer(HybridBlock):
    def __init__(self, growth_rate, bn_size, dropout, layout="NCHW",
                 **kwargs):
        super().__init__(**kwargs)
        ax = 3 if layout == "NHWC" else 1
        self._cdim = ax
        self.body = nn.HybridSequential(prefix="")
        self.body.add(nn.BatchNorm(axis=ax))
        self.body.add(nn.Activation("relu"))
        self.body.add(nn.Conv2D(bn_size * growth_rate, kernel_size=1,
                                use_bias=False, layout=layout))
        self.body.add(nn.BatchNorm(axis=ax))
        self.body.add(nn.Activation("relu"))
        self.body.add(nn.Conv2D(growth_rate, kernel_size=3, padding=1,
                                use_bias=False, layout=layout))
        if dropout:
            self.body.add(nn.Dropout(dropout))

    def forward(self, x):
        out = self.body(x)
        return F_nd.concat(x, out, dim=self._cdim)

    hybrid_forward = None


def _make_dense_block(num_layers, bn_size, growth_rate, dropout,
                      stage_index, layout="NCHW"):
    out = nn.HybridSequential(prefix=f"stage{stage_index}_")
    with out.name_scope():
        for _ in range(num_layers):
            out.add(_DenseLayer(growth_rate, bn_size, dropout,
                                layout=layout))
    return out


def _make_transition(num_output_features, layout="NCHW"):
    ax = 3 if layout == "NHWC" else 1
    out = nn.HybridSequential(prefix="")
    out.add(nn.BatchNorm(axis=ax))
    out.add(nn.Activation("relu"))
    out.add(nn.Conv2D(num_output_features, kernel_size=1, use_bias=False,
                      layout=layout))
    out.add(nn.AvgPool2D(pool_size=2, strides=2, layout=layout))
    return out


class DenseNet(HybridBlock):
    def __init__(self, num_init_features, growth_rate, block_config,
                 bn_size=4, dropout=0, classes=1000, layout="NCHW",
                 **kwargs):
        super().__init__(**kwargs)
        L = layout
        ax = 3 if L == "NHWC" else 1
        with self.name_scope():
            self.features = nn.HybridSequential(prefix="")
            self.features.add(nn.Conv2D(num_init_features, kernel_size=7,
                                        strides=2, padding=3,
                                        use_bias=False, layout=L))
            self.features.add(nn.BatchNorm(axis=ax))
            self.features.add(nn.Activation("relu"))
            self.features.add(nn.MaxPool2D(pool_size=3, strides=2,
                                           padding=1, layout=L))
            num_features = num_init_features
            for i, num_layers in enumerate(block_config):
                self.features.add(_make_dense_block(
                    num_layers, bn_size, growth_rate, dropout, i + 1,
                    layout=L))
                num_features = num_features + num_layers * growth_rate
                if i != len(block_config) - 1:
                    self.features.add(_make_transition(num_features // 2,
                                                       layout=L))
                    num_features = num_features // 2
            self.features.add(nn.BatchNorm(axis=ax))
            self.features.add(nn.Activation("relu"))
            self.features.add(nn.AvgPool2D(pool_size=7, layout=L))
            self.features.add(nn.Flatten())
            self.output = nn.Dense(classes)

    def forward(self, x):
        return self.output(self.features(x))

    hybrid_forward = None


_densenet_spec = {121: (64, 32, [6, 12, 24, 16]),
                  161: (96, 48, [6, 12, 36, 24]),
                  169: (64, 32, [6, 12, 32, 32]),
                  201: (64, 32, [6, 12, 48, 32])}


def get_densenet(num_layers, **kwargs):
    num_init_features, growth_rate, block_config = _densenet_spec[num_layers]
    return DenseNet(num_init_features, growth_rate, block_config, **kwargs)


def densenet121(**kw):
    return get_densenet(121, **kw)


def densenet161(**kw):
    return get_densenet(161, **kw)


def densenet169(**kw):
    return get_densenet(169, **kw)


def densenet201(**kw):
    return get_densenet(201, **kw)
