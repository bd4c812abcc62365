#!/usr/bin/env python3
"""Word-level LSTM language model (reference parity: example/rnn/
word_lm — the classic PTB LSTM LM; model pattern from
gluon/rnn/rnn_layer.py cells over embeddings + tied softmax).

Synthetic token stream (no network for PTB); trains with truncated BPTT
and reports perplexity.

    python examples/rnn/lm.py --steps 20
"""
import argparse
import math
import os
import sys
import time

import numpy as np

sys.path.insert(0, os.path.join(os.path.dirname(os.path.abspath(__file__)),
                                "..", ".."))
import anand_mxnet_amd as mx  # noqa: E402
from anand_mxnet_amd import autograd, gluon, nd  # noqa: E402
from anand_mxnet_amd.gluon import nn, rnn  # noqa: E402


class RNNModel(gluon.Block):
    def __init__(self, vocab, embed=200, hidden=200, layers=2,
                 dropout=0.2, **kw):
        super().__init__(**kw)
        with self.name_scope():
            self.drop = nn.Dropout(dropout)
            self.encoder = nn.Embedding(vocab, embed)
            self.rnn = rnn.LSTM(hidden, num_layers=layers,
                                layout="TNC", input_size=embed)
            self.decoder = nn.Dense(vocab, in_units=hidden,
                                    flatten=False)

    def forward(self, inputs, states):
        emb = self.drop(self.encoder(inputs))     # [T, N, E]
        out, states = self.rnn(emb, states)
        out = self.drop(out)
        return self.decoder(out), states


def detach(states):
    if isinstance(states, (list, tuple)):
        return [detach(s) for s in states]
    return states.detach()


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--steps", type=int, default=20)
    ap.add_argument("--batch-size", type=int, default=16)
    ap.add_argument("--bptt", type=int, default=35)
    ap.add_argument("--vocab", type=int, default=2000)
    ap.add_argument("--clip", type=float, default=0.25)
    ap.add_argument("--lr", type=float, default=1.0)
    ap.add_argument("--log-interval", type=int, default=5)
    args = ap.parse_args()

    use_gpu = mx.context.num_gpus() > 0
    ctx = mx.gpu(0) if use_gpu else mx.cpu()

    model = RNNModel(args.vocab)
    model.initialize(mx.init.Xavier(), ctx=ctx)
    trainer = gluon.Trainer(model.collect_params(), "sgd",
                            {"learning_rate": args.lr,
                             "clip_gradient": args.clip})
    loss_fn = gluon.loss.SoftmaxCrossEntropyLoss()

    rng = np.random.RandomState(0)
    # one long synthetic token stream, sliced into BPTT windows
    stream = rng.randint(0, args.vocab,
                         (args.steps + 1) * args.bptt + 1)
    states = model.rnn.begin_state(args.batch_size, ctx=ctx)
    total_L, total_tok = 0.0, 0
    t0 = time.time()
    for step in range(1, args.steps + 1):
        lo = (step - 1) * args.bptt
        window = stream[lo:lo + args.bptt + 1]
        data = np.tile(window[:-1, None], (1, args.batch_size))
        target = np.tile(window[1:, None], (1, args.batch_size))
        x = nd.array(data, ctx=ctx, dtype="int64")
        y = nd.array(target, ctx=ctx, dtype="int64")
        states = detach(states)
        with autograd.record():
            out, states = model(x, states)
            L = loss_fn(out.reshape(-1, args.vocab), y.reshape(-1))
            L = L.mean()
        L.backward()
        trainer.step(1)
        total_L += float(L.asscalar()) * args.bptt * args.batch_size
        total_tok += args.bptt * args.batch_size
        if step % args.log_interval == 0 or step == args.steps:
            avg = total_L / total_tok
            ppl = math.exp(min(avg, 20))
            tok_s = total_tok / (time.time() - t0)
            print(f"step {step:4d}  loss {avg:.3f}  "
                  f"ppl {ppl:,.1f}  {tok_s:,.0f} tok/s")


if __name__ == "__main__":
    main()
