#!/usr/bin/env python3
"""Transformer-big translation training example (reference parity:
example/machine-translation / the Sockeye-style WMT setup the reference
benchmarks; model: gluon/model_zoo/transformer.py).

Synthetic token data (this image has no network for WMT downloads);
demonstrates the full train loop: label-smoothed CE, Adam with the
inverse-sqrt warmup schedule, gradient clipping, checkpointing.

    python examples/machine-translation/train.py --steps 30 [--fp8]
"""
import argparse
import os
import sys
import time

import numpy as np

sys.path.insert(0, os.path.join(os.path.dirname(os.path.abspath(__file__)),
                                "..", ".."))
import anand_mxnet_amd as mx  # noqa: E402
from anand_mxnet_amd import autograd, gluon, nd  # noqa: E402
from anand_mxnet_amd.gluon.model_zoo.transformer import (  # noqa: E402
    transformer_big)


class WarmupInvSqrt:
    """Attention-is-all-you-need schedule: d^-0.5 * min(t^-0.5,
    t * warmup^-1.5)."""

    def __init__(self, d_model, warmup):
        self.scale = d_model ** -0.5
        self.warmup = warmup

    def __call__(self, t):
        t = max(t, 1)
        return self.scale * min(t ** -0.5, t * self.warmup ** -1.5)


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--steps", type=int, default=30)
    ap.add_argument("--batch-size", type=int, default=16)
    ap.add_argument("--seq-len", type=int, default=512)
    ap.add_argument("--vocab", type=int, default=32768)
    ap.add_argument("--warmup-steps", type=int, default=4000)
    ap.add_argument("--label-smoothing", type=float, default=0.1)
    ap.add_argument("--clip", type=float, default=1.0)
    ap.add_argument("--fp8", action="store_true",
                    help="MX block-scaled fp8 projections (MI355X)")
    ap.add_argument("--log-interval", type=int, default=5)
    ap.add_argument("--save-prefix", default=None)
    args = ap.parse_args()

    use_gpu = mx.context.num_gpus() > 0
    ctx = mx.gpu(0) if use_gpu else mx.cpu()
    B, T, V = args.batch_size, args.seq_len, args.vocab
    if not use_gpu:  # keep the CPU demo snappy
        B, T, V = 2, 64, 1024

    net = transformer_big(src_vocab=V, tgt_vocab=V, use_fp8=args.fp8)
    if use_gpu:
        net.cast("bfloat16")
    net.initialize(mx.init.Xavier(magnitude=2.34), ctx=ctx)

    sched = WarmupInvSqrt(1024, args.warmup_steps)
    trainer = gluon.Trainer(net.collect_params(), "adam",
                            {"learning_rate": sched(1), "beta1": 0.9,
                             "beta2": 0.98, "epsilon": 1e-9,
                             "clip_gradient": args.clip})
    loss_fn = gluon.loss.SoftmaxCrossEntropyLoss()

    rng = np.random.RandomState(0)
    src = nd.array(rng.randint(1, V, (B, T)), ctx=ctx, dtype="int64")
    tgt = nd.array(rng.randint(1, V, (B, T)), ctx=ctx, dtype="int64")
    eps = args.label_smoothing

    t0 = time.time()
    for step in range(1, args.steps + 1):
        trainer.set_learning_rate(sched(step))
        with autograd.record():
            logits = net(src, tgt)  # [B, T, V]
            flat = logits.astype("float32").reshape(-1, V)
            lsm = nd.log_softmax(flat, axis=-1)
            nll = -nd.pick(lsm, tgt.reshape(-1), axis=1)
            smooth = -lsm.mean(axis=1)
            loss = ((1 - eps) * nll + eps * smooth).mean()
        loss.backward()
        trainer.step(1)
        if step % args.log_interval == 0 or step == args.steps:
            tok_s = B * T * step / (time.time() - t0)
            print(f"step {step:4d}  loss {float(loss.asscalar()):.4f}  "
                  f"lr {sched(step):.2e}  {tok_s:,.0f} tok/s")

    if args.save_prefix:
        net.save_parameters(f"{args.save_prefix}.params")
        trainer.save_states(f"{args.save_prefix}.states")
        print(f"saved {args.save_prefix}.params/.states")


if __name__ == "__main__":
    main()
