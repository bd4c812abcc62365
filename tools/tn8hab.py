#!/usr/bin/env python3
"""A/B the 256x128 window TN kernel (MXA_TN8H) on short-K grids."""
import json
import os
import subprocess
import sys

SHAPES = [(8192, 3072, 768), (16384, 768, 768), (16384, 2304, 768),
          (8192, 768, 768), (16384, 3072, 768)]

CHILD = r"""
import json, os, sys, time
import torch
sys.path.insert(0, ".")
from anand_mxnet_amd import _C
BF = torch.bfloat16
DEV = torch.device("cuda", 0)
out = {"arm": os.environ.get("MXA_TN8H", "0")}
torch.manual_seed(0)
a = (torch.rand((8192, 768), device=DEV) * 2 - 1).to(BF)
b = (torch.rand((3072, 768), device=DEV) * 2 - 1).to(BF)
y = _C.gemm_tn(a, b, None, False)
ref = a.float() @ b.float().t()
out["relerr"] = round(((y.float() - ref).abs().max() /
                       ref.abs().max()).item(), 5)
res = []
for (M, N, K) in json.loads(os.environ["AB_SHAPES"]):
    a = (torch.rand((M, K), device=DEV) * 2 - 1).to(BF)
    b = (torch.rand((N, K), device=DEV) * 2 - 1).to(BF)
    for _ in range(5):
        _C.gemm_tn(a, b, None, False)
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(20):
        _C.gemm_tn(a, b, None, False)
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
        env["MXA_TN8H"] = arm
        r = subprocess.run([sys.executable, "-c", CHILD], env=env,
                           capture_output=True, text=True,
                           cwd=os.path.dirname(os.path.dirname(
                               os.path.abspath(__file__))))
        print(r.stdout.strip())
        if r.returncode != 0:
            print("ARM", arm, "FAILED:", r.stderr[-1500:])


if __name__ == "__main__":
    main()
