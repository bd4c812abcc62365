#!/usr/bin/env python3
"""NLP benchmarks (BASELINE.json configs #4/#5):
  --model bert_base       : BERT-base MLM+NSP pretrain step, bf16, seq 512
  --model transformer_big : WMT transformer-big training step

Same contract as bench.py (one JSON line from rank 0; N ranks over RCCL)."""
import argparse
import json
import os
import sys
import time

import numpy as np
import torch
import torch.distributed as dist

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--model", default="bert_base",
                    choices=["bert_base", "transformer_big"])
    ap.add_argument("--steps", type=int, default=10)
    ap.add_argument("--warmup", type=int, default=3)
    ap.add_argument("--batch-size", type=int, default=32,
                    help="per-GPU sequences")
    ap.add_argument("--seq-len", type=int, default=512)
    ap.add_argument("--dtype", default="bfloat16")
    ap.add_argument("--fp8", action="store_true",
                    help="transformer_big only: fp8 e4m3 MFMA projections")
    ap.add_argument("--capture", action="store_true",
                    help="hipGraph whole-step capture")
    args = ap.parse_args()

    world_size = int(os.environ.get("WORLD_SIZE", "1"))
    rank = int(os.environ.get("RANK", "0"))
    local_rank = int(os.environ.get("LOCAL_RANK", "0"))
    local_rank = local_rank % max(torch.cuda.device_count(), 1)
    distributed = world_size > 1
    if distributed:
        os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
        torch.cuda.set_device(local_rank)
        dist.init_process_group(
            backend=os.environ.get("MXA_DIST_BACKEND", "nccl"))
        try:
            t = torch.tensor([1.0], device=f"cuda:{local_rank}")
            dist.all_reduce(t)
            torch.cuda.synchronize()
        except Exception:
            dist.destroy_process_group()
            dist.init_process_group(backend="gloo")

    import anand_mxnet_amd as mx
    from anand_mxnet_amd import gluon, autograd, nd
    from anand_mxnet_amd import _C

    ctx = mx.gpu(local_rank)
    torch.cuda.set_device(local_rank)
    rng = np.random.RandomState(99 + rank)
    B, T = args.batch_size, args.seq_len
    loss_fn = gluon.loss.SoftmaxCrossEntropyLoss()

    if args.model == "bert_base":
        from anand_mxnet_amd.gluon.model_zoo.bert import bert_base, \
            BERTForPretrain
        vocab = 30528
        net = BERTForPretrain(bert_base(vocab_size=vocab, max_length=T),
                              vocab_size=vocab)
        net.cast(args.dtype)
        net.initialize(mx.init.Normal(0.02), ctx=ctx)
        trainer = gluon.Trainer(net.collect_params(), "adam",
                                {"learning_rate": 1e-4,
                                 "multi_precision": True})
        tok = nd.array(rng.randint(0, vocab, (B, T)), ctx=ctx,
                       dtype="int64")
        tt = nd.zeros((B, T), ctx=ctx, dtype="int64")
        mlm_label = nd.array(rng.randint(0, vocab, (B, T)), ctx=ctx,
                             dtype="int64")
        nsp_label = nd.array(rng.randint(0, 2, (B,)), ctx=ctx,
                             dtype="int64")

        def step():
            _C.rng_step(local_rank)  # graph-safe dropout seed advance
            with autograd.record():
                mlm, nsp = net(tok, tt)
                l = loss_fn(mlm.reshape(-1, vocab),
                            mlm_label.reshape(-1)).mean() + \
                    loss_fn(nsp, nsp_label).mean()
            l.backward()
            trainer.step(B * world_size)
            return l
        tokens_per_step = B * T * world_size
        cfg_model = "bert_base"
    else:
        from anand_mxnet_amd.gluon.model_zoo.transformer import \
            transformer_big
        vocab = 32768
        net = transformer_big(src_vocab=vocab, tgt_vocab=vocab,
                              max_length=max(T, 1024), use_fp8=args.fp8)
        net.cast(args.dtype)
        net.initialize(mx.init.Normal(0.02), ctx=ctx)
        trainer = gluon.Trainer(net.collect_params(), "adam",
                                {"learning_rate": 1e-4,
                                 "multi_precision": True})
        src = nd.array(rng.randint(0, vocab, (B, T)), ctx=ctx,
                       dtype="int64")
        tgt = nd.array(rng.randint(0, vocab, (B, T)), ctx=ctx,
                       dtype="int64")

        def step():
            _C.rng_step(local_rank)  # graph-safe dropout seed advance
            with autograd.record():
                out = net(src, tgt)
                l = loss_fn(out.reshape(-1, vocab),
                            tgt.reshape(-1)).mean()
            l.backward()
            trainer.step(B * world_size)
            return l
        tokens_per_step = B * T * world_size
        cfg_model = "transformer_big_fp8" if args.fp8 else "transformer_big"

    for _ in range(args.warmup):
        loss = step()
    torch.cuda.synchronize()
    graph = None
    if args.capture:
        try:
            g = torch.cuda.CUDAGraph()
            with torch.cuda.graph(g):
                loss = step()
            g.replay()
            torch.cuda.synchronize()
            graph = g
        except Exception as e:
            print(f"# graph capture unavailable ({e}); eager",
                  file=sys.stderr)
    run_step = graph.replay if graph is not None else step
    for _ in range(2):
        run_step()
    torch.cuda.synchronize()
    if distributed:
        dist.barrier()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(args.steps):
        out = run_step()
        if out is not None:  # eager path returns the loss; graph replay
            loss = out       # updates the captured static loss in place

    torch.cuda.synchronize()
    if distributed:
        dist.barrier()
    torch.cuda.synchronize()
    elapsed = time.perf_counter() - t0
    if distributed:
        t = torch.tensor([elapsed], device=f"cuda:{local_rank}")
        dist.all_reduce(t, op=dist.ReduceOp.MAX)
        elapsed = t.item()

    if rank == 0:
        print(json.dumps({
            "metric": f"tokens/sec (whole node) {cfg_model} "
                      f"{args.dtype} seq{T}",
            "value": round(tokens_per_step * args.steps / elapsed, 1),
            "unit": "tokens/sec",
            "n_gpus": world_size,
            "steps": args.steps,
            "warmup": args.warmup,
            "ms_per_step": round(elapsed / args.steps * 1000, 3),
            "higher_is_better": True,
            "scaling": "weak",
            "vs_baseline": None,
            "dtype": args.dtype,
            "data": "synthetic",
            "config": {"model": cfg_model,
                       "global_batch": B * world_size, "seq_len": T,
                       "parallelism": f"dp{world_size}",
                       "loss": float(loss.asscalar())},
        }))
    if distributed:
        dist.destroy_process_group()


if __name__ == "__main__":
    main()
