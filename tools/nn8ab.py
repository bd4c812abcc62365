#!/usr/bin/env python3
"""A/B the window NN GEMM (MXA_NN8G) vs gemm_nn_core."""
import json
import os
import subprocess
import sys

SHAPES = [(8192, 768, 3072), (8192, 3072, 768), (4096, 4096, 4096),
          (16384, 1024, 4096), (16384, 4096, 1024), (8192, 8192, 8192),
          (200704, 2048, 512), (8192, 768, 768)]

CHILD = r"""
import json, os, sys, time
import torch
sys.path.insert(0, ".")
from anand_mxnet_amd import _C
BF = torch.bfloat16
DEV = torch.device("cuda", 0)
out = {"arm": os.environ.get("MXA_NN8G", "0")}
torch.manual_seed(0)
# correctness: 4096x4096x4096 + a tail shape vs fp32
for (M, K, N) in [(4096, 4096, 4096), (1000, 512, 520)]:
    a = (torch.rand((M, K), device=DEV) * 2 - 1).to(BF)
    b = (torch.rand((K, N), device=DEV) * 2 - 1).to(BF)
    y = _C.gemm_nn(a, b)
    ref = a.float() @ b.float()
    err = (y.float() - ref).abs().max().item() / ref.abs().max().item()
    out[f"relerr_{M}x{N}x{K}"] = round(err, 5)
res = []
for (M, K, N) in json.loads(os.environ["AB_SHAPES"]):
    a = (torch.rand((M, K), device=DEV) * 2 - 1).to(BF)
    b = (torch.rand((K, N), device=DEV) * 2 - 1).to(BF)
    for _ in range(5):
        _C.gemm_nn(a, b)
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(20):
        _C.gemm_nn(a, b)
    torch.cuda.synchronize()
    t = (time.perf_counter() - t0) / 20
    res.append({"shape": f"{M}x{N}x{K}",
                "TF": round(2.0 * M * N * K / t / 1e12, 1)})
out["gemms"] = res
print(json.dumps(out))
"""


def main():
    env = dict(os.environ)
    env["AB_SHAPES"] = json.dumps(SHAPES)
    for arm in ("0", "1"):
        env["MXA_NN8G"] = arm
        r = subprocess.run([sys.executable, "-c", CHILD], env=env,
                           capture_output=True, text=True,
                           cwd=os.path.dirname(os.path.dirname(
                               os.path.abspath(__file__))))
        print(r.stdout.strip())
        if r.returncode != 0:
            print("ARM", arm, "FAILED:", r.stderr[-1500:])


if __name__ == "__main__":
    main()
