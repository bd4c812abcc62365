#!/usr/bin/env python3
"""Single-kernel probe for PMC counter runs (keeps dispatch count small)."""
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
from anand_mxnet_amd import _C

BF = torch.bfloat16
DEV = torch.device("cuda", 0)

which = sys.argv[1] if len(sys.argv) > 1 else "wg"
iters = int(sys.argv[2]) if len(sys.argv) > 2 else 3

if which == "wg":
    m, Mo, No = 100352, 128, 128
    a = (torch.rand((m, Mo), device=DEV) * 2 - 1).to(BF)
    b = (torch.rand((m, No), device=DEV) * 2 - 1).to(BF)
    for _ in range(iters):
        _C.gemm_wg(a, b)
elif which == "wg_conv":
    x = (torch.rand((64, 28, 28, 128), device=DEV) * 2 - 1).to(BF)
    dy = (torch.rand((64, 28, 28, 128), device=DEV) * 2 - 1).to(BF)
    for _ in range(iters):
        _C.conv2d_wgrad(x, dy, [3, 3], [1, 1], [1, 1], [1, 1])
elif which == "tn":
    a = (torch.rand((8192, 8192), device=DEV) * 2 - 1).to(BF)
    b = (torch.rand((8192, 8192), device=DEV) * 2 - 1).to(BF)
    for _ in range(iters):
        _C.gemm_tn(a, b, None, False)
elif which == "flash":
    B, H, T, D = 32, 12, 512, 64
    q = (torch.rand(B, T, H*D, device=DEV)*2-1).to(BF)
    k = (torch.rand(B, T, H*D, device=DEV)*2-1).to(BF)
    v = (torch.rand(B, T, H*D, device=DEV)*2-1).to(BF)
    sc = 1.0/(D**0.5)
    for _ in range(iters):
        out, m, l = _C.flash_fwd(q, k, v, H, sc, False, 0.0, 7)
        dq, dk, dv = _C.flash_bwd(q, k, v, out, q, m, l, H, sc, False,
                                  0.0, 7)
elif which == "nn":
    a = (torch.rand((8192, 8192), device=DEV) * 2 - 1).to(BF)
    b = (torch.rand((8192, 8192), device=DEV) * 2 - 1).to(BF)
    for _ in range(iters):
        _C.gemm_nn(a, b)
torch.cuda.synchronize()
print("done", which)
