"""Gluon convolution / pooling layers (parity with
python/mxnet/gluon/nn/conv_layers.py).

MI355X-native note: layout='NHWC' is the performance layout (channels-last;
the implicit-GEMM CDNA4 conv kernel is NHWC-native; C-contiguous rows feed
MFMA K-slices directly).  NCHW is supported for reference parity; weight
shape follows data layout: NCHW->(K,C,R,S), NHWC->(K,R,S,C)."""
from __future__ import annotations

import numpy as np

from ..block import HybridBlock
from .basic_layers import Activation

__all__ = ["Conv2D", "Conv1D", "Conv2DTranspose", "MaxPool2D", "AvgPool2D",
           "GlobalAvgPool2D", "GlobalMaxPool2D", "MaxPool1D", "AvgPool1D",
           "Conv3D", "Conv1DTranspose", "MaxPool3D", "AvgPool3D",
           "GlobalMaxPool1D", "GlobalAvgPool1D", "GlobalMaxPool3D",
           "GlobalAvgPool3D", "ReflectionPad2D"]


def _pair(x):
    if isinstance(x, (list, tuple)):
        return tuple(x)
    return (x, x)


class Conv2D(HybridBlock):
    def __init__(self, channels, kernel_size, strides=(1, 1), padding=(0, 0),
                 dilation=(1, 1), groups=1, layout="NCHW", activation=None,
                 use_bias=True, weight_initializer=None,
                 bias_initializer="zeros", in_channels=0, prefix=None,
                 params=None):
        super().__init__(prefix=prefix, params=params)
        with self.name_scope():
            self._channels = channels
            self._in_channels = in_channels
            self._kernel = _pair(kernel_size)
            self._strides = _pair(strides)
            self._padding = _pair(padding)
            self._dilation = _pair(dilation)
            self._groups = groups
            self._layout = layout
            kr, ks = self._kernel
            if layout == "NCHW":
                wshape = (channels, in_channels // groups if in_channels else 0,
                          kr, ks)
            elif layout == "NHWC":
                wshape = (channels, kr, ks,
                          in_channels // groups if in_channels else 0)
            else:
                raise ValueError(f"unsupported layout {layout}")
            self.weight = self.params.get("weight", shape=wshape,
                                          init=weight_initializer,
                                          allow_deferred_init=True)
            if use_bias:
                self.bias = self.params.get("bias", shape=(channels,),
                                            init=bias_initializer,
                                            allow_deferred_init=True)
            else:
                self.bias = None
            self.act = Activation(activation) if activation else None

    def shape_infer(self, x, *args):
        c_axis = 1 if self._layout == "NCHW" else 3
        in_c = x.shape[c_axis]
        kr, ks = self._kernel
        if self._layout == "NCHW":
            self.weight.shape = (self._channels, in_c // self._groups, kr, ks)
        else:
            self.weight.shape = (self._channels, kr, ks, in_c // self._groups)
        if self.bias is not None:
            self.bias.shape = (self._channels,)

    def hybrid_forward(self, F, x, weight, bias=None):
        if bias is None and getattr(self, "_folded_bias", None) is not None:
            bias = self._folded_bias  # from utils.fold_batchnorm
        out = F.Convolution(x, weight, bias, kernel=self._kernel,
                            stride=self._strides, dilate=self._dilation,
                            pad=self._padding, num_filter=self._channels,
                            num_group=self._groups, no_bias=bias is None,
                            layout=self._layout,
                            collect_bn_stats=getattr(
                                self, "_collect_bn_stats", False),
                            fold_relu=getattr(self, "_fold_relu", False))
        if self.act is not None:
            out = self.act(out)
        return out


class Conv1D(HybridBlock):
    def __init__(self, channels, kernel_size, strides=1, padding=0,
                 dilation=1, groups=1, activation=None, use_bias=True,
                 weight_initializer=None, bias_initializer="zeros",
                 in_channels=0, **kwargs):
        super().__init__(**kwargs)
        with self.name_scope():
            self._channels = channels
            self._kernel = kernel_size if isinstance(kernel_size, int) \
                else kernel_size[0]
            self._strides = strides if isinstance(strides, int) else strides[0]
            self._padding = padding if isinstance(padding, int) else padding[0]
            self._dilation = dilation if isinstance(dilation, int) else dilation[0]
            self._groups = groups
            self.weight = self.params.get(
                "weight", shape=(channels,
                                 in_channels // groups if in_channels else 0,
                                 self._kernel),
                init=weight_initializer, allow_deferred_init=True)
            self.bias = self.params.get("bias", shape=(channels,),
                                        init=bias_initializer,
                                        allow_deferred_init=True) \
                if use_bias else None
            self.act = Activation(activation) if activation else None

    def shape_infer(self, x, *args):
        self.weight.shape = (self._channels, x.shape[1] // self._groups,
                             self._kernel)
        if self.bias is not None:
            self.bias.shape = (self._channels,)

    def hybrid_forward(self, F, x, weight, bias=None):
        # express as 2D conv with height 1
        x4 = x.expand_dims(2)
        w4 = weight.expand_dims(2)
        out = F.Convolution(x4, w4, bias, kernel=(1, self._kernel),
                            stride=(1, self._strides),
                            dilate=(1, self._dilation),
                            pad=(0, self._padding),
                            num_filter=self._channels,
                            num_group=self._groups, no_bias=bias is None,
                            layout="NCHW")
        out = out.squeeze(2)
        if self.act is not None:
            out = self.act(out)
        return out


class Conv2DTranspose(HybridBlock):
    def __init__(self, channels, kernel_size, strides=(1, 1), padding=(0, 0),
                 output_padding=(0, 0), dilation=(1, 1), groups=1,
                 layout="NCHW", activation=None, use_bias=True,
                 weight_initializer=None, bias_initializer="zeros",
                 in_channels=0, **kwargs):
        super().__init__(**kwargs)
        with self.name_scope():
            self._channels = channels
            self._kernel = _pair(kernel_size)
            self._strides = _pair(strides)
            self._padding = _pair(padding)
            self._output_padding = _pair(output_padding)
            self._dilation = _pair(dilation)
            self._groups = groups
            self.weight = self.params.get(
                "weight", shape=(in_channels if in_channels else 0,
                                 channels // groups) + self._kernel,
                init=weight_initializer, allow_deferred_init=True)
            self.bias = self.params.get("bias", shape=(channels,),
                                        init=bias_initializer,
                                        allow_deferred_init=True) \
                if use_bias else None
            self.act = Activation(activation) if activation else None

    def shape_infer(self, x, *args):
        self.weight.shape = (x.shape[1], self._channels // self._groups) + \
            self._kernel
        if self.bias is not None:
            self.bias.shape = (self._channels,)

    def hybrid_forward(self, F, x, weight, bias=None):
        out = F.Deconvolution(x, weight, bias, kernel=self._kernel,
                              stride=self._strides, dilate=self._dilation,
                              pad=self._padding, adj=self._output_padding,
                              num_filter=self._channels,
                              num_group=self._groups, no_bias=bias is None)
        if self.act is not None:
            out = self.act(out)
        return out


class _Pool2D(HybridBlock):
    _pool_type = "max"

    def __init__(self, pool_size=(2, 2), strides=None, padding=0,
                 layout="NCHW", ceil_mode=False, count_include_pad=True,
                 **kwargs):
        super().__init__(**kwargs)
        self._kwargs = {
            "kernel": _pair(pool_size),
            "stride": _pair(strides) if strides is not None else _pair(pool_size),
            "pad": _pair(padding),
            "layout": layout,
            "pool_type": self._pool_type,
            "pooling_convention": "full" if ceil_mode else "valid",
            "count_include_pad": count_include_pad,
        }

    def hybrid_forward(self, F, x):
        return F.Pooling(x, **self._kwargs)


class MaxPool2D(_Pool2D):
    _pool_type = "max"


class AvgPool2D(_Pool2D):
    _pool_type = "avg"


class MaxPool1D(HybridBlock):
    def __init__(self, pool_size=2, strides=None, padding=0, **kwargs):
        super().__init__(**kwargs)
        self._k = pool_size
        self._s = strides or pool_size
        self._p = padding

    def hybrid_forward(self, F, x):
        x4 = x.expand_dims(2)
        out = F.Pooling(x4, kernel=(1, self._k), stride=(1, self._s),
                        pad=(0, self._p), pool_type="max")
        return out.squeeze(2)


class AvgPool1D(HybridBlock):
    def __init__(self, pool_size=2, strides=None, padding=0, **kwargs):
        super().__init__(**kwargs)
        self._k = pool_size
        self._s = strides or pool_size
        self._p = padding

    def hybrid_forward(self, F, x):
        x4 = x.expand_dims(2)
        out = F.Pooling(x4, kernel=(1, self._k), stride=(1, self._s),
                        pad=(0, self._p), pool_type="avg")
        return out.squeeze(2)


class GlobalAvgPool2D(HybridBlock):
    def __init__(self, layout="NCHW", **kwargs):
        super().__init__(**kwargs)
        self._layout = layout

    def hybrid_forward(self, F, x):
        return F.Pooling(x, global_pool=True, pool_type="avg",
                         layout=self._layout, kernel=(1, 1))


class GlobalMaxPool2D(HybridBlock):
    def __init__(self, layout="NCHW", **kwargs):
        super().__init__(**kwargs)
        self._layout = layout

    def hybrid_forward(self, F, x):
        return F.Pooling(x, global_pool=True, pool_type="max",
                         layout=self._layout, kernel=(1, 1))


class Conv3D(HybridBlock):
    """3-D convolution, NCDHW (reference conv_layers.py — Conv3D)."""

    def __init__(self, channels, kernel_size, strides=(1, 1, 1),
                 padding=(0, 0, 0), dilation=(1, 1, 1), groups=1,
                 layout="NCDHW", activation=None, use_bias=True,
                 weight_initializer=None, bias_initializer="zeros",
                 in_channels=0, **kwargs):
        super().__init__(**kwargs)
        assert layout == "NCDHW", "Conv3D: NCDHW only"
        with self.name_scope():
            self._channels = channels
            self._kernel = _triple(kernel_size)
            self._strides = _triple(strides)
            self._padding = _triple(padding)
            self._dilation = _triple(dilation)
            self._groups = groups
            self.weight = self.params.get(
                "weight", shape=(channels,
                                 in_channels // groups if in_channels else 0)
                + self._kernel,
                init=weight_initializer, allow_deferred_init=True)
            self.bias = self.params.get("bias", shape=(channels,),
                                        init=bias_initializer,
                                        allow_deferred_init=True) \
                if use_bias else None
            self.act = Activation(activation) if activation else None

    def shape_infer(self, x, *args):
        self.weight.shape = (self._channels,
                             x.shape[1] // self._groups) + self._kernel
        if self.bias is not None:
            self.bias.shape = (self._channels,)

    def hybrid_forward(self, F, x, weight, bias=None):
        out = F.Convolution(x, weight, bias, kernel=self._kernel,
                            stride=self._strides, dilate=self._dilation,
                            pad=self._padding, num_filter=self._channels,
                            num_group=self._groups, no_bias=bias is None,
                            layout="NCDHW")
        if self.act is not None:
            out = self.act(out)
        return out


class Conv1DTranspose(HybridBlock):
    """1-D transposed conv (reference conv_layers.py)."""

    def __init__(self, channels, kernel_size, strides=1, padding=0,
                 output_padding=0, dilation=1, groups=1, use_bias=True,
                 weight_initializer=None, bias_initializer="zeros",
                 in_channels=0, **kwargs):
        super().__init__(**kwargs)
        with self.name_scope():
            self._channels = channels
            k = kernel_size if isinstance(kernel_size, int) \
                else kernel_size[0]
            self._kernel = (k,)
            self._strides = (strides if isinstance(strides, int)
                             else strides[0],)
            self._padding = (padding if isinstance(padding, int)
                             else padding[0],)
            self._adj = (output_padding if isinstance(output_padding, int)
                         else output_padding[0],)
            self._dilation = (dilation if isinstance(dilation, int)
                              else dilation[0],)
            self._groups = groups
            self.weight = self.params.get(
                "weight",
                shape=(in_channels, channels // groups) + self._kernel
                if in_channels else (0, channels // groups) + self._kernel,
                init=weight_initializer, allow_deferred_init=True)
            self.bias = self.params.get("bias", shape=(channels,),
                                        init=bias_initializer,
                                        allow_deferred_init=True) \
                if use_bias else None

    def shape_infer(self, x, *args):
        self.weight.shape = (x.shape[1],
                             self._channels // self._groups) + self._kernel
        if self.bias is not None:
            self.bias.shape = (self._channels,)

    def hybrid_forward(self, F, x, weight, bias=None):
        return F.Deconvolution(x, weight, bias, kernel=self._kernel,
                               stride=self._strides, dilate=self._dilation,
                               pad=self._padding, adj=self._adj,
                               num_filter=self._channels,
                               num_group=self._groups,
                               no_bias=bias is None)


class MaxPool3D(HybridBlock):
    def __init__(self, pool_size=(2, 2, 2), strides=None, padding=0,
                 ceil_mode=False, **kwargs):
        super().__init__(**kwargs)
        self._k = _triple(pool_size)
        self._s = _triple(strides) if strides is not None else self._k
        self._p = _triple(padding)
        self._conv = "full" if ceil_mode else "valid"

    def hybrid_forward(self, F, x):
        return F.Pooling(x, kernel=self._k, stride=self._s, pad=self._p,
                         pool_type="max", pooling_convention=self._conv)


class AvgPool3D(MaxPool3D):
    def hybrid_forward(self, F, x):
        return F.Pooling(x, kernel=self._k, stride=self._s, pad=self._p,
                         pool_type="avg", pooling_convention=self._conv)


class GlobalMaxPool1D(HybridBlock):
    def hybrid_forward(self, F, x):
        return F.Pooling(x, kernel=(1,), global_pool=True, pool_type="max")


class GlobalAvgPool1D(HybridBlock):
    def hybrid_forward(self, F, x):
        return F.Pooling(x, kernel=(1,), global_pool=True, pool_type="avg")


class GlobalMaxPool3D(HybridBlock):
    def hybrid_forward(self, F, x):
        return F.Pooling(x, kernel=(1, 1, 1), global_pool=True,
                         pool_type="max")


class GlobalAvgPool3D(HybridBlock):
    def hybrid_forward(self, F, x):
        return F.Pooling(x, kernel=(1, 1, 1), global_pool=True,
                         pool_type="avg")


class ReflectionPad2D(HybridBlock):
    """Reflection padding on H/W of NCHW (reference conv_layers.py)."""

    def __init__(self, padding=0, **kwargs):
        super().__init__(**kwargs)
        self._pad = padding if isinstance(padding, (list, tuple)) \
            else (padding,) * 4

    def hybrid_forward(self, F, x):
        l, r, t, b = (int(p) for p in self._pad)
        return F.pad(x, mode="reflect",
                     pad_width=(0, 0, 0, 0, t, b, l, r))


def _triple(v):
    return (v, v, v) if isinstance(v, int) else tuple(v)
