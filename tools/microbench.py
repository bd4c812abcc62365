#!/usr/bin/env python3
"""Kernel microbenchmarks on MI355X: TF/s for the MFMA GEMM/conv kernels,
GB/s for the memory-bound kernels.  Run under gpurun; summary is committed
to profiles/ as measurement evidence."""
import argparse
import json
import time

import os
import sys

import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
from anand_mxnet_amd import _C

BF = torch.bfloat16
DEV = torch.device("cuda", 0)


def timeit(fn, iters=20, warmup=5):
    for _ in range(warmup):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters


def bench_gemm(M, N, K):
    a = (torch.rand((M, K), device=DEV) * 2 - 1).to(BF)
    b = (torch.rand((N, K), device=DEV) * 2 - 1).to(BF)
    t = timeit(lambda: _C.gemm_tn(a, b, None, False))
    tf = 2.0 * M * N * K / t / 1e12
    tt = timeit(lambda: torch.matmul(a, b.t()))
    tf_blas = 2.0 * M * N * K / tt / 1e12
    return {"op": f"gemm_tn {M}x{N}x{K}", "ours_TF": round(tf, 1),
            "rocblas_TF": round(tf_blas, 1), "ms": round(t * 1e3, 3)}


def bench_mx(M, N, K):
    a = (torch.rand((M, K), device=DEV) * 2 - 1).to(BF)
    b = (torch.rand((N, K), device=DEV) * 2 - 1).to(BF)
    a8, asc = _C.mx_cast(a.contiguous())
    b8, bsc = _C.mx_cast(b.contiguous())
    t = timeit(lambda: _C.gemm_mx(a8, asc, b8, bsc))
    tf = 2.0 * M * N * K / t / 1e12
    tc = timeit(lambda: _C.mx_cast(a.contiguous()))
    return {"op": f"gemm_mx_fp8 {M}x{N}x{K}", "ours_TF": round(tf, 1),
            "ms": round(t * 1e3, 3), "cast_ms": round(tc * 1e3, 3)}


def bench_wg(m, Mo, No):
    a = (torch.rand((m, Mo), device=DEV) * 2 - 1).to(BF)
    b = (torch.rand((m, No), device=DEV) * 2 - 1).to(BF)
    t = timeit(lambda: _C.gemm_wg(a, b))
    tf = 2.0 * m * Mo * No / t / 1e12
    return {"op": f"gemm_wg m={m} {Mo}x{No}", "ours_TF": round(tf, 1),
            "ms": round(t * 1e3, 3)}


def bench_conv(N, H, W, C, K, R, s, p, name=""):
    x = (torch.rand((N, H, W, C), device=DEV) * 2 - 1).to(BF)
    w = (torch.rand((R, R, K, C), device=DEV) * 0.1).to(BF)
    P = (H + 2 * p - R) // s + 1
    flops = 2.0 * N * P * P * K * R * R * C
    t = timeit(lambda: _C.conv2d_fwd(x, w, None, [s, s], [p, p], [1, 1],
                                     False))
    out = {"op": f"conv {name} N{N} {H}x{W}x{C}->K{K} r{R}s{s}",
           "fwd_TF": round(flops / t / 1e12, 1), "fwd_ms": round(t * 1e3, 3)}
    # dgrad
    dy = (torch.rand((N, P, P, K), device=DEV) * 2 - 1).to(BF)
    w2 = (torch.rand((R, R, C, K), device=DEV) * 0.1).to(BF)
    td = timeit(lambda: _C.conv2d_dgrad(dy, w2, [N, H, W, C], [s, s],
                                        [p, p], [1, 1]))
    out["dgrad_TF"] = round(flops / td / 1e12, 1)
    tw = timeit(lambda: _C.conv2d_wgrad(x, dy, [R, R], [s, s], [p, p],
                                        [1, 1]))
    out["wgrad_TF"] = round(flops / tw / 1e12, 1)
    return out


def bench_bn(M, C):
    x = torch.rand((M, C), device=DEV).to(BF).reshape(M // 64, 8, 8, C)
    g = torch.ones(C, device=DEV)
    b = torch.zeros(C, device=DEV)
    rm = torch.zeros(C, device=DEV)
    rv = torch.ones(C, device=DEV)
    t = timeit(lambda: _C.bn_fwd_train(x, g, b, rm, rv, 1e-5, 0.9, True))
    bytes_ = 2 * 2 * M * C  # read + write bf16 (stats pass reads again)
    return {"op": f"bn_fwd_train M={M} C={C}",
            "GBps": round(bytes_ * 1.5 / t / 1e9, 0),
            "ms": round(t * 1e3, 3)}


def bench_softmax(rows, D):
    x = (torch.rand((rows, D), device=DEV) * 4).to(BF)
    t = timeit(lambda: _C.softmax_fwd(x, False))
    return {"op": f"softmax {rows}x{D}",
            "GBps": round(2 * 2 * rows * D / t / 1e9, 0),
            "ms": round(t * 1e3, 3)}


def bench_conv_wg(N, H, W, C, K, R, s, p, name=""):
    x = (torch.rand((N, H, W, C), device=DEV) * 2 - 1).to(BF)
    P = (H + 2 * p - R) // s + 1
    flops = 2.0 * N * P * P * K * R * R * C
    dy = (torch.rand((N, P, P, K), device=DEV) * 2 - 1).to(BF)
    tw = timeit(lambda: _C.conv2d_wgrad(x, dy, [R, R], [s, s], [p, p],
                                        [1, 1]))
    return {"op": f"convwg {name} N{N} {H}x{W}x{C}->K{K} r{R}s{s}",
            "wgrad_TF": round(flops / tw / 1e12, 1)}


def main():
    results = []
    if "--mx-only" in sys.argv:
        for shape in [(4096, 4096, 4096), (8192, 8192, 8192),
                      (8192, 4096, 1024), (8192, 1024, 4096)]:
            print(json.dumps(bench_mx(*shape)))
        return
    if "--nn-only" in sys.argv:
        for (M, K, N) in [(8192, 768, 768), (8192, 768, 3072),
                          (8192, 3072, 768), (4096, 4096, 4096),
                          (16384, 1024, 4096), (16384, 4096, 1024),
                          (8192, 8192, 8192), (200704, 2048, 512)]:
            a = (torch.rand((M, K), device=DEV) * 2 - 1).to(BF)
            b = (torch.rand((K, N), device=DEV) * 2 - 1).to(BF)
            t = timeit(lambda: _C.gemm_nn(a, b))
            print(json.dumps({"op": f"gemm_nn {M}x{N}x{K}",
                              "ours_TF": round(2.0 * M * N * K / t / 1e12, 1),
                              "ms": round(t * 1e3, 3)}))
        return
    if "--conv-only" in sys.argv:
        for cfg in [(64, 56, 56, 64, 64, 3, 1, 1, "res2-3x3"),
                    (64, 56, 56, 64, 256, 1, 1, 0, "res2-1x1"),
                    (64, 56, 56, 256, 64, 1, 1, 0, "res2-red"),
                    (64, 28, 28, 128, 128, 3, 1, 1, "res3-3x3"),
                    (64, 28, 28, 512, 128, 1, 1, 0, "res3-red"),
                    (64, 14, 14, 256, 256, 3, 1, 1, "res4-3x3"),
                    (64, 14, 14, 1024, 256, 1, 1, 0, "res4-red"),
                    (64, 7, 7, 512, 512, 3, 1, 1, "res5-3x3"),
                    (64, 7, 7, 512, 2048, 1, 1, 0, "res5-exp")]:
            print(json.dumps(bench_conv(*cfg)))
        return
    if "--wg-only" in sys.argv:
        results.append(bench_wg(100352, 128, 128))
        results.append(bench_wg(25088, 512, 256))
        results.append(bench_wg(8192, 4096, 4096))
        results.append(bench_conv_wg(64, 56, 56, 64, 64, 3, 1, 1, "res2-3x3"))
        results.append(bench_conv_wg(64, 56, 56, 64, 256, 1, 1, 0, "res2-1x1"))
        results.append(bench_conv_wg(64, 28, 28, 128, 128, 3, 1, 1, "res3-3x3"))
        results.append(bench_conv_wg(64, 14, 14, 256, 256, 3, 1, 1, "res4-3x3"))
        results.append(bench_conv_wg(64, 7, 7, 512, 512, 3, 1, 1, "res5-3x3"))
        for r in results:
            print(json.dumps(r))
        return
    for shape in [(4096, 4096, 4096), (8192, 8192, 8192),
                  (12544, 256, 1152), (3136, 512, 2304), (25088, 128, 576),
                  (8192, 768, 768), (8192, 3072, 768), (8192, 768, 3072)]:
        results.append(bench_gemm(*shape))
    results.append(bench_wg(100352, 128, 128))
    results.append(bench_wg(25088, 512, 256))
    # ResNet-50 v1.5 batch-64 representative layers
    results.append(bench_conv(64, 56, 56, 64, 64, 3, 1, 1, "res2-3x3"))
    results.append(bench_conv(64, 56, 56, 64, 256, 1, 1, 0, "res2-1x1"))
    results.append(bench_conv(64, 28, 28, 128, 128, 3, 1, 1, "res3-3x3"))
    results.append(bench_conv(64, 14, 14, 256, 256, 3, 1, 1, "res4-3x3"))
    results.append(bench_conv(64, 7, 7, 512, 512, 3, 1, 1, "res5-3x3"))
    results.append(bench_bn(64 * 56 * 56, 256))
    results.append(bench_softmax(16 * 512, 512))
    for r in results:
        print(json.dumps(r))


if __name__ == "__main__":
    main()
