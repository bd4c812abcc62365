"""ResNet v1 / v1.5 / v2 (parity with gluon/model_zoo/vision/resnet.py).

MI355X-native: ``layout='NHWC'`` is the performance layout — every conv/BN/
pool runs the NHWC CDNA4 kernels with no transposes; the flagship benchmark
(BASELINE.json: ResNet-50 v1.5 bf16) uses it.  ``resnet50_v1_5`` puts the
stride in the 3x3 (the config the baseline names; upstream calls this
variant resnet50_v1b / "ResNet50 v1.5").
"""
from __future__ import annotations

from ... import nn
from ...block import HybridBlock

__all__ = ["ResNetV1", "ResNetV2", "BasicBlockV1", "BasicBlockV2",
           "BottleneckV1", "BottleneckV2", "resnet18_v1", "resnet34_v1",
           "resnet50_v1", "resnet101_v1", "resnet152_v1", "resnet18_v2",
           "resnet34_v2", "resnet50_v2", "resnet101_v2", "resnet152_v2",
           "resnet50_v1_5", "get_resnet"]


def _bn_axis(layout):
    return 3 if layout == "NHWC" else 1


def _conv3x3(channels, stride, in_channels, layout):
    return nn.Conv2D(channels, kernel_size=3, strides=stride, padding=1,
                     use_bias=False, in_channels=in_channels, layout=layout)


class BasicBlockV1(HybridBlock):
    def __init__(self, channels, stride, downsample=False, in_channels=0,
                 layout="NCHW", **kwargs):
        super().__init__(**kwargs)
        ax = _bn_axis(layout)
        self.body = nn.HybridSequential(prefix="")
        self.body.add(_conv3x3(channels, stride, in_channels, layout))
        self.body.add(nn.BatchNorm(axis=ax, fuse_relu=True))
        self.body.add(_conv3x3(channels, 1, channels, layout))
        # junction BN: relu(bn(conv_out) + residual) fused in one kernel
        self.bn_join = nn.BatchNorm(axis=ax)
        if downsample:
            self.downsample = nn.HybridSequential(prefix="")
            self.downsample.add(nn.Conv2D(channels, kernel_size=1,
                                          strides=stride, use_bias=False,
                                          in_channels=in_channels,
                                          layout=layout))
            self.downsample.add(nn.BatchNorm(axis=ax))
        else:
            self.downsample = None

    def forward(self, x):
        residual = x
        x_out = self.body(x)
        if self.downsample is not None:
            residual = self.downsample(residual)
        return self.bn_join(x_out, residual)

    hybrid_forward = None


class BottleneckV1(HybridBlock):
    """v1: stride in first 1x1; v1.5 (stride_in_3x3): stride in the 3x3.

    BN+ReLU pairs use the fused-ReLU BatchNorm (CDNA4 kernel epilogue —
    one less full-tensor round trip per pair)."""

    def __init__(self, channels, stride, downsample=False, in_channels=0,
                 layout="NCHW", stride_in_3x3=False, **kwargs):
        super().__init__(**kwargs)
        ax = _bn_axis(layout)
        s1, s3 = (1, stride) if stride_in_3x3 else (stride, 1)
        self.body = nn.HybridSequential(prefix="")
        self.body.add(nn.Conv2D(channels // 4, kernel_size=1, strides=s1,
                                use_bias=False, layout=layout))
        self.body.add(nn.BatchNorm(axis=ax, fuse_relu=True))
        self.body.add(_conv3x3(channels // 4, s3, channels // 4, layout))
        self.body.add(nn.BatchNorm(axis=ax, fuse_relu=True))
        self.body.add(nn.Conv2D(channels, kernel_size=1, strides=1,
                                use_bias=False, layout=layout))
        # junction BN: relu(bn(conv_out) + residual) fused in one kernel
        self.bn_join = nn.BatchNorm(axis=ax)
        if downsample:
            self.downsample = nn.HybridSequential(prefix="")
            self.downsample.add(nn.Conv2D(channels, kernel_size=1,
                                          strides=stride, use_bias=False,
                                          in_channels=in_channels,
                                          layout=layout))
            self.downsample.add(nn.BatchNorm(axis=ax))
        else:
            self.downsample = None

    def forward(self, x):
        residual = x
        x_out = self.body(x)
        if self.downsample is not None:
            residual = self.downsample(residual)
        return self.bn_join(x_out, residual)

    hybrid_forward = None


class BasicBlockV2(HybridBlock):
    def __init__(self, channels, stride, downsample=False, in_channels=0,
                 layout="NCHW", **kwargs):
        super().__init__(**kwargs)
        ax = _bn_axis(layout)
        self.bn1 = nn.BatchNorm(axis=ax)
        self.conv1 = _conv3x3(channels, stride, in_channels, layout)
        self.bn2 = nn.BatchNorm(axis=ax)
        self.conv2 = _conv3x3(channels, 1, channels, layout)
        if downsample:
            self.downsample = nn.Conv2D(channels, 1, stride, use_bias=False,
                                        in_channels=in_channels,
                                        layout=layout)
        else:
            self.downsample = None

    def forward(self, x):
        from .... import ndarray as F
        residual = x
        x = self.bn1(x)
        x = F.Activation(x, act_type="relu")
        if self.downsample is not None:
            residual = self.downsample(x)
        x = self.conv1(x)
        x = self.bn2(x)
        x = F.Activation(x, act_type="relu")
        x = self.conv2(x)
        return x + residual

    hybrid_forward = None


class BottleneckV2(HybridBlock):
    def __init__(self, channels, stride, downsample=False, in_channels=0,
                 layout="NCHW", **kwargs):
        super().__init__(**kwargs)
        ax = _bn_axis(layout)
        self.bn1 = nn.BatchNorm(axis=ax)
        self.conv1 = nn.Conv2D(channels // 4, 1, 1, use_bias=False,
                               layout=layout)
        self.bn2 = nn.BatchNorm(axis=ax)
        self.conv2 = _conv3x3(channels // 4, stride, channels // 4, layout)
        self.bn3 = nn.BatchNorm(axis=ax)
        self.conv3 = nn.Conv2D(channels, 1, 1, use_bias=False, layout=layout)
        if downsample:
            self.downsample = nn.Conv2D(channels, 1, stride, use_bias=False,
                                        in_channels=in_channels,
                                        layout=layout)
        else:
            self.downsample = None

    def forward(self, x):
        from .... import ndarray as F
        residual = x
        x = self.bn1(x)
        x = F.Activation(x, act_type="relu")
        if self.downsample is not None:
            residual = self.downsample(x)
        x = self.conv1(x)
        x = self.bn2(x)
        x = F.Activation(x, act_type="relu")
        x = self.conv2(x)
        x = self.bn3(x)
        x = F.Activation(x, act_type="relu")
        x = self.conv3(x)
        return x + residual

    hybrid_forward = None


class ResNetV1(HybridBlock):
    def __init__(self, block, layers, channels, classes=1000, layout="NCHW",
                 thumbnail=False, stride_in_3x3=False, **kwargs):
        super().__init__(**kwargs)
        assert len(layers) == len(channels) - 1
        ax = _bn_axis(layout)
        self._layout = layout
        with self.name_scope():
            self.features = nn.HybridSequential(prefix="")
            if thumbnail:
                self.features.add(_conv3x3(channels[0], 1, 0, layout))
            else:
                self.features.add(nn.Conv2D(channels[0], 7, 2, 3,
                                            use_bias=False, layout=layout))
                self.features.add(nn.BatchNorm(axis=ax, fuse_relu=True))
                self.features.add(nn.MaxPool2D(3, 2, 1, layout=layout))
            for i, num_layer in enumerate(layers):
                stride = 1 if i == 0 else 2
                self.features.add(self._make_layer(
                    block, num_layer, channels[i + 1], stride, i + 1,
                    in_channels=channels[i], layout=layout,
                    stride_in_3x3=stride_in_3x3))
            self.features.add(nn.GlobalAvgPool2D(layout=layout))
            self.output = nn.Dense(classes, in_units=channels[-1])

    def _make_layer(self, block, layers, channels, stride, stage_index,
                    in_channels=0, layout="NCHW", stride_in_3x3=False):
        extra = {"stride_in_3x3": stride_in_3x3} \
            if block is BottleneckV1 else {}
        layer = nn.HybridSequential(prefix=f"stage{stage_index}_")
        with layer.name_scope():
            layer.add(block(channels, stride,
                            channels != in_channels, in_channels=in_channels,
                            layout=layout, prefix="", **extra))
            for _ in range(layers - 1):
                layer.add(block(channels, 1, False, in_channels=channels,
                                layout=layout, prefix="", **extra))
        return layer

    def forward(self, x):
        x = self.features(x)
        x = self.output(x)
        return x

    hybrid_forward = None


class ResNetV2(HybridBlock):
    def __init__(self, block, layers, channels, classes=1000, layout="NCHW",
                 thumbnail=False, **kwargs):
        super().__init__(**kwargs)
        assert len(layers) == len(channels) - 1
        ax = _bn_axis(layout)
        self._layout = layout
        with self.name_scope():
            self.features = nn.HybridSequential(prefix="")
            self.features.add(nn.BatchNorm(axis=ax, scale=False,
                                           center=False))
            if thumbnail:
                self.features.add(_conv3x3(channels[0], 1, 0, layout))
            else:
                self.features.add(nn.Conv2D(channels[0], 7, 2, 3,
                                            use_bias=False, layout=layout))
                self.features.add(nn.BatchNorm(axis=ax))
                self.features.add(nn.Activation("relu"))
                self.features.add(nn.MaxPool2D(3, 2, 1, layout=layout))
            in_channels = channels[0]
            for i, num_layer in enumerate(layers):
                stride = 1 if i == 0 else 2
                self.features.add(self._make_layer(
                    block, num_layer, channels[i + 1], stride, i + 1,
                    in_channels=in_channels, layout=layout))
                in_channels = channels[i + 1]
            self.features.add(nn.BatchNorm(axis=ax))
            self.features.add(nn.Activation("relu"))
            self.features.add(nn.GlobalAvgPool2D(layout=layout))
            self.features.add(nn.Flatten())
            self.output = nn.Dense(classes, in_units=in_channels)

    def _make_layer(self, block, layers, channels, stride, stage_index,
                    in_channels=0, layout="NCHW"):
        layer = nn.HybridSequential(prefix=f"stage{stage_index}_")
        with layer.name_scope():
            layer.add(block(channels, stride, channels != in_channels,
                            in_channels=in_channels, layout=layout,
                            prefix=""))
            for _ in range(layers - 1):
                layer.add(block(channels, 1, False, in_channels=channels,
                                layout=layout, prefix=""))
        return layer

    def forward(self, x):
        x = self.features(x)
        x = self.output(x)
        return x

    hybrid_forward = None


_resnet_spec = {
    18: ("basic_block", [2, 2, 2, 2], [64, 64, 128, 256, 512]),
    34: ("basic_block", [3, 4, 6, 3], [64, 64, 128, 256, 512]),
    50: ("bottle_neck", [3, 4, 6, 3], [64, 256, 512, 1024, 2048]),
    101: ("bottle_neck", [3, 4, 23, 3], [64, 256, 512, 1024, 2048]),
    152: ("bottle_neck", [3, 8, 36, 3], [64, 256, 512, 1024, 2048]),
}
_resnet_block_versions = [
    {"basic_block": BasicBlockV1, "bottle_neck": BottleneckV1},
    {"basic_block": BasicBlockV2, "bottle_neck": BottleneckV2},
]
_resnet_net_versions = [ResNetV1, ResNetV2]


def get_resnet(version, num_layers, pretrained=False, ctx=None,
               layout="NCHW", stride_in_3x3=False, **kwargs):
    block_type, layers, channels = _resnet_spec[num_layers]
    resnet_class = _resnet_net_versions[version - 1]
    block_class = _resnet_block_versions[version - 1][block_type]
    extra = {"stride_in_3x3": stride_in_3x3} if version == 1 else {}
    net = resnet_class(block_class, layers, channels, layout=layout,
                       **extra, **kwargs)
    if version == 1:
        _mark_conv_bn_stats(net)
    if pretrained:
        raise RuntimeError("no network access: pretrained weights "
                           "unavailable; load explicit .params instead")
    return net


def _mark_conv_bn_stats(block):
    """In v1 nets every conv output feeds a BatchNorm, so ask the conv
    kernels to fold the BN per-channel sum/sumsq into their GEMM epilogue
    (ops/hip conv2d_nhwc collect_stats; the BN picks the sums up from the
    tensor and skips its stats pass)."""
    for child in block._children.values():
        if isinstance(child, nn.Conv2D):
            child._collect_bn_stats = True
        else:
            _mark_conv_bn_stats(child)


def resnet18_v1(**kw):
    return get_resnet(1, 18, **kw)


def resnet34_v1(**kw):
    return get_resnet(1, 34, **kw)


def resnet50_v1(**kw):
    return get_resnet(1, 50, **kw)


def resnet50_v1_5(**kw):
    """ResNet-50 v1.5 (stride in 3x3) — the BASELINE.json flagship config."""
    return get_resnet(1, 50, stride_in_3x3=True, **kw)


def resnet101_v1(**kw):
    return get_resnet(1, 101, **kw)


def resnet152_v1(**kw):
    return get_resnet(1, 152, **kw)


def resnet18_v2(**kw):
    return get_resnet(2, 18, **kw)


def resnet34_v2(**kw):
    return get_resnet(2, 34, **kw)


def resnet50_v2(**kw):
    return get_resnet(2, 50, **kw)


def resnet101_v2(**kw):
    return get_resnet(2, 101, **kw)


def resnet152_v2(**kw):
    return get_resnet(2, 152, **kw)
