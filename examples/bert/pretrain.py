#!/usr/bin/env python3
"""BERT pretraining example (parity with GluonNLP's run_pretraining.py
surface at the scale of this repo): masked-LM + next-sentence-prediction
objective on synthetic data (offline environment), Gluon training loop,
bf16, fused flash attention.

  python examples/bert/pretrain.py --gpus 0 --batch-size 16 --steps 50
  python examples/bert/pretrain.py            # CPU smoke (tiny model)
"""
import argparse
import logging
import os
import sys
import time

import numpy as np

sys.path.insert(0, os.path.join(os.path.dirname(os.path.abspath(__file__)),
                                "..", ".."))

import anand_mxnet_amd as mx
from anand_mxnet_amd import autograd, gluon, nd
from anand_mxnet_amd.gluon.model_zoo.bert import (BERTForPretrain,
                                                  bert_base, bert_tiny)


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--gpus", default="", help="e.g. '0'; empty = CPU")
    ap.add_argument("--model", default=None,
                    help="bert_base (default on GPU) or bert_tiny")
    ap.add_argument("--batch-size", type=int, default=None)
    ap.add_argument("--seq-len", type=int, default=None)
    ap.add_argument("--steps", type=int, default=20)
    ap.add_argument("--lr", type=float, default=1e-4)
    ap.add_argument("--log-interval", type=int, default=10)
    args = ap.parse_args()

    logging.basicConfig(level=logging.INFO)
    use_gpu = args.gpus != ""
    ctx = mx.gpu(int(args.gpus.split(",")[0])) if use_gpu else mx.cpu()
    model = args.model or ("bert_base" if use_gpu else "bert_tiny")
    B = args.batch_size or (16 if use_gpu else 2)
    T = args.seq_len or (512 if use_gpu else 128)

    vocab = 30528
    backbone = (bert_base if model == "bert_base" else bert_tiny)(
        vocab_size=vocab, max_length=T)
    net = BERTForPretrain(backbone, vocab_size=vocab)
    if use_gpu:
        net.cast("bfloat16")
    net.initialize(mx.init.Normal(0.02), ctx=ctx)
    trainer = gluon.Trainer(net.collect_params(), "adam",
                            {"learning_rate": args.lr,
                             "multi_precision": use_gpu})
    loss_fn = gluon.loss.SoftmaxCrossEntropyLoss()

    rng = np.random.RandomState(0)
    tok = nd.array(rng.randint(0, vocab, (B, T)), ctx=ctx, dtype="int64")
    tt = nd.zeros((B, T), ctx=ctx, dtype="int64")
    mlm_y = nd.array(rng.randint(0, vocab, (B, T)), ctx=ctx, dtype="int64")
    nsp_y = nd.array(rng.randint(0, 2, (B,)), ctx=ctx, dtype="int64")

    t0 = time.perf_counter()
    for step in range(1, args.steps + 1):
        with autograd.record():
            mlm, nsp = net(tok, tt)
            loss = loss_fn(mlm.reshape(-1, vocab),
                           mlm_y.reshape(-1)).mean() + \
                loss_fn(nsp, nsp_y).mean()
        loss.backward()
        trainer.step(B)
        if step % args.log_interval == 0 or step == args.steps:
            dt = time.perf_counter() - t0
            tput = step * B * T / dt
            logging.info("step %d  loss %.4f  %.0f tokens/sec", step,
                         float(loss.asscalar()), tput)


if __name__ == "__main__":
    main()
