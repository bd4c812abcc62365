#!/usr/bin/env python3
"""Inference throughput (reference parity:
example/image-classification/benchmark_score.py): images/sec per model
at several batch sizes, forward-only under autograd.pause, bf16 NHWC.

  gpurun -- 'python tools/benchmark_score.py'
"""
import os
import sys
import time

import numpy as np

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import anand_mxnet_amd as mx
from anand_mxnet_amd import nd
from anand_mxnet_amd.gluon.model_zoo import vision as zoo
import torch


def score(name, factory, batch_sizes, size=224, layout="NHWC",
          fold=None):
    if fold is None:
        fold = "--fold-bn" in sys.argv
    ctx = mx.gpu(0)
    net = factory()
    net.cast("bfloat16")
    net.initialize(mx.init.Xavier(), ctx=ctx)
    if fold:
        from anand_mxnet_amd.gluon.utils import fold_batchnorm
        shape0 = (1, size, size, 3) if layout == "NHWC" else (1, 3, size,
                                                              size)
        net(nd.zeros(shape0, dtype="bfloat16", ctx=ctx))  # materialize
        n = fold_batchnorm(net)
        name = f"{name} +fold{n}"
    for b in batch_sizes:
        shape = (b, size, size, 3) if layout == "NHWC" else (b, 3, size,
                                                             size)
        x = nd.random.uniform(-1, 1, shape=shape, dtype="bfloat16",
                              ctx=ctx)
        for _ in range(5):
            net(x)
        torch.cuda.synchronize()
        t0 = time.perf_counter()
        iters = 20
        for _ in range(iters):
            net(x)
        torch.cuda.synchronize()
        dt = time.perf_counter() - t0
        print(f"{name:24s} batch {b:4d}: "
              f"{b * iters / dt:10.1f} images/sec")


def main():
    score("resnet50_v1.5", lambda: zoo.resnet50_v1_5(layout="NHWC"),
          [1, 32, 128, 256])
    score("resnet18_v1", lambda: zoo.resnet18_v1(layout="NHWC"), [1, 64])
    score("resnet18_v1 (NCHW)", lambda: zoo.resnet18_v1(), [1, 64],
          layout="NCHW")
    score("vgg11", lambda: zoo.vgg11(layout="NHWC"), [1, 64])
    score("vgg11 (NCHW)", lambda: zoo.vgg11(), [1, 64], layout="NCHW")
    score("mobilenet1_0", lambda: zoo.mobilenet1_0(layout="NHWC"),
          [1, 64])
    score("squeezenet1_1", lambda: zoo.squeezenet1_1(layout="NHWC"),
          [1, 64])
    score("densenet121", lambda: zoo.densenet121(layout="NHWC"), [64])


if __name__ == "__main__":
    main()
