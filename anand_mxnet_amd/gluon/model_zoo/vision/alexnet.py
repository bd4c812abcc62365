"""AlexNet (parity with gluon/model_zoo/vision/alexnet.py)."""
from ... import nn
from ...block import HybridBlock

__all__ = ["AlexNet", "alexnet"]


class AlexNet(HybridBlock):
    def __init__(self, classes=1000, layout="NCHW", **kwargs):
        super().__init__(**kwargs)
        L = layout
        with self.name_scope():
            self.features = nn.HybridSequential(prefix="")
            with self.features.name_scope():
                self.features.add(nn.Conv2D(64, kernel_size=11, strides=4,
                                            padding=2, activation="relu", layout=L))
                self.features.add(nn.MaxPool2D(pool_size=3, strides=2, layout=L))
                self.features.add(nn.Conv2D(192, kernel_size=5, padding=2,
                                            activation="relu", layout=L))
                self.features.add(nn.MaxPool2D(pool_size=3, strides=2, layout=L))
                self.features.add(nn.Conv2D(384, kernel_size=3, padding=1,
                                            activation="relu", layout=L))
                self.features.add(nn.Conv2D(256, kernel_size=3, padding=1,
                                            activation="relu", layout=L))
                self.features.add(nn.Conv2D(256, kernel_size=3, padding=1,
                                            activation="relu", layout=L))
                self.features.add(nn.MaxPool2D(pool_size=3, strides=2, layout=L))
                self.features.add(nn.Flatten())
                self.features.add(nn.Dense(4096, activation="relu"))
                self.features.add(nn.Dropout(0.5))
                self.features.add(nn.Dense(4096, activation="relu"))
                self.features.add(nn.Dropout(0.5))
            self.output = nn.Dense(classes)

    def forward(self, x):
        return self.output(self.features(x))

    hybrid_forward = None


def alexnet(**kwargs):
    return AlexNet(**kwargs)
