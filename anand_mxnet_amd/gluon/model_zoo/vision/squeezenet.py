"""SqueezeNet (parity with gluon/model_zoo/vision/squeezenet.py)."""
from ... import nn
from ...block import HybridBlock
from .... import ndarray as F_nd

__all__ = ["SqueezeNet", "squeezenet1_0", "squeezenet1_1"]


class _Fire(HybridBlock):
    def __init__(self, squeeze_channels, expand1x1_channels,
                 expand3x3_channels, layout="NCHW", **kwargs):
        super().__init__(**kwargs)
        self._cdim = 3 if layout == "NHWC" else 1
        self.squeeze = nn.Conv2D(squeeze_channels, kernel_size=1,
                                 activation="relu", layout=layout)
        self.expand1x1 = nn.Conv2D(expand1x1_channels, kernel_size=1,
                                   activation="relu", layout=layout)
        self.expand3x3 = nn.Conv2D(expand3x3_channels, kernel_size=3,
                                   padding=1, activation="relu",
                                   layout=layout)

    def forward(self, x):
        x = self.squeeze(x)
        return F_nd.concat(self.expand1x1(x), self.expand3x3(x),
                           dim=self._cdim)

    hybrid_forward = None


class SqueezeNet(HybridBlock):
    def __init__(self, version, classes=1000, layout="NCHW", **kwargs):
        super().__init__(**kwargs)
        assert version in ("1.0", "1.1")
        L = layout
        with self.name_scope():
            self.features = nn.HybridSequential(prefix="")
            if version == "1.0":
                self.features.add(nn.Conv2D(96, kernel_size=7, strides=2,
                                            activation="relu", layout=L))
                self.features.add(nn.MaxPool2D(3, 2, ceil_mode=True, layout=L))
                self.features.add(_Fire(16, 64, 64, layout=L))
                self.features.add(_Fire(16, 64, 64, layout=L))
                self.features.add(_Fire(32, 128, 128, layout=L))
                self.features.add(nn.MaxPool2D(3, 2, ceil_mode=True, layout=L))
                self.features.add(_Fire(32, 128, 128, layout=L))
                self.features.add(_Fire(48, 192, 192, layout=L))
                self.features.add(_Fire(48, 192, 192, layout=L))
                self.features.add(_Fire(64, 256, 256, layout=L))
                self.features.add(nn.MaxPool2D(3, 2, ceil_mode=True, layout=L))
                self.features.add(_Fire(64, 256, 256, layout=L))
            else:
                self.features.add(nn.Conv2D(64, kernel_size=3, strides=2,
                                            activation="relu", layout=L))
                self.features.add(nn.MaxPool2D(3, 2, ceil_mode=True, layout=L))
                self.features.add(_Fire(16, 64, 64, layout=L))
                self.features.add(_Fire(16, 64, 64, layout=L))
                self.features.add(nn.MaxPool2D(3, 2, ceil_mode=True, layout=L))
                self.features.add(_Fire(32, 128, 128, layout=L))
                self.features.add(_Fire(32, 128, 128, layout=L))
                self.features.add(nn.MaxPool2D(3, 2, ceil_mode=True, layout=L))
                self.features.add(_Fire(48, 192, 192, layout=L))
                self.features.add(_Fire(48, 192, 192, layout=L))
                self.features.add(_Fire(64, 256, 256, layout=L))
                self.features.add(_Fire(64, 256, 256, layout=L))
            self.features.add(nn.Dropout(0.5))
            self.output = nn.HybridSequential(prefix="")
            self.output.add(nn.Conv2D(classes, kernel_size=1,
                                      activation="relu", layout=L))
            self.output.add(nn.GlobalAvgPool2D(layout=L))
            self.output.add(nn.Flatten())

    def forward(self, x):
        return self.output(self.features(x))

    hybrid_forward = None


def squeezenet1_0(**kw):
    return SqueezeNet("1.0", **kw)


def squeezenet1_1(**kw):
    return SqueezeNet("1.1", **kw)
