#!/usr/bin/env python3
"""Canonical training CLI (parity with the reference's
example/image-classification/common/fit.py argument surface:
--network --gpus --kv-store --batch-size --lr --lr-step-epochs --dtype
--load-epoch --model-prefix ...), Gluon training loop with Speedometer.

Data: --data synthetic (default, no-network environment) or --data mnist
with raw idx files under --data-dir.
"""
import argparse
import logging
import os
import sys

import numpy as np

sys.path.insert(0, os.path.join(os.path.dirname(os.path.abspath(__file__)),
                                "..", ".."))

import anand_mxnet_amd as mx
from anand_mxnet_amd import gluon, nd, autograd
from anand_mxnet_amd.callback import Speedometer, BatchEndParam
from anand_mxnet_amd.gluon.model_zoo import vision as zoo
from anand_mxnet_amd.io import NDArrayIter
from anand_mxnet_amd.lr_scheduler import MultiFactorScheduler

logging.basicConfig(level=logging.INFO)


def lenet(classes=10):
    net = gluon.nn.HybridSequential(prefix="lenet_")
    with net.name_scope():
        net.add(gluon.nn.Conv2D(20, 5, activation="relu"),
                gluon.nn.MaxPool2D(2, 2),
                gluon.nn.Conv2D(50, 5, activation="relu"),
                gluon.nn.MaxPool2D(2, 2),
                gluon.nn.Flatten(),
                gluon.nn.Dense(500, activation="relu"),
                gluon.nn.Dense(classes))
    return net


def get_net(name, classes, layout):
    if name == "lenet":
        return lenet(classes)
    kw = {"classes": classes}
    if "resnet" in name:
        kw["layout"] = layout
    return zoo.get_model(name, **kw)


def get_data(args):
    if args.data == "mnist":
        from anand_mxnet_amd.io import MNISTIter
        train = MNISTIter(os.path.join(args.data_dir,
                                       "train-images-idx3-ubyte"),
                          os.path.join(args.data_dir,
                                       "train-labels-idx1-ubyte"),
                          batch_size=args.batch_size)
        val = MNISTIter(os.path.join(args.data_dir,
                                     "t10k-images-idx3-ubyte"),
                        os.path.join(args.data_dir,
                                     "t10k-labels-idx1-ubyte"),
                        batch_size=args.batch_size, shuffle=False)
        return train, val, (1, 28, 28)
    # synthetic
    shape = tuple(int(s) for s in args.image_shape.split(","))
    n = args.num_examples
    if args.layout == "NHWC":
        shape = (shape[1], shape[2], shape[0])
    x = np.random.uniform(-1, 1, (n,) + shape).astype(np.float32)
    y = np.random.randint(0, args.num_classes, (n,)).astype(np.float32)
    train = NDArrayIter(x, y, batch_size=args.batch_size, shuffle=True)
    return train, None, shape


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--network", default="resnet50_v1_5")
    ap.add_argument("--gpus", default="",
                    help="e.g. '0' or '0,1'; empty = cpu")
    ap.add_argument("--kv-store", default="device")
    ap.add_argument("--batch-size", type=int, default=128)
    ap.add_argument("--num-epochs", type=int, default=1)
    ap.add_argument("--lr", type=float, default=0.1)
    ap.add_argument("--lr-step-epochs", default="30,60,80")
    ap.add_argument("--lr-factor", type=float, default=0.1)
    ap.add_argument("--mom", type=float, default=0.9)
    ap.add_argument("--wd", type=float, default=1e-4)
    ap.add_argument("--dtype", default="float32",
                    choices=["float32", "bfloat16", "float16"])
    ap.add_argument("--optimizer", default="sgd")
    ap.add_argument("--data", default="synthetic",
                    choices=["synthetic", "mnist"])
    ap.add_argument("--data-dir", default="data/")
    ap.add_argument("--image-shape", default="3,224,224")
    ap.add_argument("--layout", default="NCHW", choices=["NCHW", "NHWC"])
    ap.add_argument("--num-classes", type=int, default=1000)
    ap.add_argument("--num-examples", type=int, default=5120)
    ap.add_argument("--disp-batches", type=int, default=20)
    ap.add_argument("--model-prefix", default=None)
    ap.add_argument("--load-epoch", type=int, default=None)
    ap.add_argument("--max-batches", type=int, default=None)
    args = ap.parse_args()

    ctx = [mx.gpu(int(i)) for i in args.gpus.split(",") if i != ""] or \
        [mx.cpu()]
    net = get_net(args.network, args.num_classes, args.layout)
    if args.dtype != "float32":
        net.cast(args.dtype)
    net.initialize(mx.init.Xavier(), ctx=ctx)
    if args.load_epoch is not None and args.model_prefix:
        net.load_parameters(f"{args.model_prefix}-{args.load_epoch:04d}"
                            ".params", ctx=ctx)

    train, val, _ = get_data(args)
    epoch_size = args.num_examples // args.batch_size
    steps = [int(e) * epoch_size for e in args.lr_step_epochs.split(",")]
    sched = MultiFactorScheduler(steps, args.lr_factor, base_lr=args.lr)
    trainer = gluon.Trainer(
        net.collect_params(), args.optimizer,
        {"learning_rate": args.lr, "momentum": args.mom, "wd": args.wd,
         "lr_scheduler": sched,
         "multi_precision": args.dtype != "float32"},
        kvstore=args.kv_store)
    loss_fn = gluon.loss.SoftmaxCrossEntropyLoss()
    metric = mx.metric.Accuracy()
    speedometer = Speedometer(args.batch_size, args.disp_batches)

    from anand_mxnet_amd.gluon.utils import split_and_load
    for epoch in range(args.num_epochs):
        train.reset()
        metric.reset()
        for nbatch, batch in enumerate(train):
            if args.max_batches and nbatch >= args.max_batches:
                break
            data = split_and_load(batch.data[0].astype(args.dtype), ctx)
            label = split_and_load(batch.label[0], ctx)
            losses, outs = [], []
            with autograd.record():
                for xd, yd in zip(data, label):
                    out = net(xd)
                    losses.append(loss_fn(out, yd))
                    outs.append(out)
            for l in losses:
                l.backward()
            trainer.step(args.batch_size)
            metric.update(label, outs)
            speedometer(BatchEndParam(epoch=epoch, nbatch=nbatch,
                                      eval_metric=metric, locals=None))
        name, acc = metric.get()
        logging.info("Epoch[%d] Train-%s=%f", epoch, name, acc)
        if args.model_prefix:
            net.save_parameters(f"{args.model_prefix}-{epoch + 1:04d}"
                                ".params")


if __name__ == "__main__":
    main()
