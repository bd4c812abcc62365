#!/usr/bin/env python3
"""A/B the guide-faithful 8-phase GEMM variant (MXA_TN8G) against the
baseline gemm_tn8_core.  Each arm runs in a subprocess (the env gate is
read once per process).  Run under gpurun; prints one JSON line per arm.
"""
import json
import os
import subprocess
import sys

SHAPES = [(4096, 4096, 4096), (8192, 8192, 8192), (8192, 8192, 4096),
          (4096, 4096, 12288)]

CHILD = r"""
import json, os, sys, time
import torch
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(
    __file__)))) if "__file__" in dir() else sys.path.insert(0, ".")
from anand_mxnet_amd import _C
BF = torch.bfloat16
DEV = torch.device("cuda", 0)
out = {"arm": os.environ.get("ARM32", os.environ.get("MXA_TN8G", "0"))}
# correctness at a shape the tn8 router actually takes (>=240 tiles of
# 256^2 and K>=1536): 4096^3 vs fp32 matmul of the same bf16 inputs
torch.manual_seed(0)
a = (torch.rand((4096, 4096), device=DEV) * 2 - 1).to(BF)
b = (torch.rand((4096, 4096), device=DEV) * 2 - 1).to(BF)
y = _C.gemm_tn(a, b, None, False)
ref = a.float() @ b.float().t()
err = (y.float() - ref).abs().max().item() / ref.abs().max().item()
out["relerr_4096"] = round(err, 5)
del ref
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
    arms = ["0", "1"]
    if os.environ.get("AB_G32"):
        arms.append("32")
    for arm in arms:
        env["MXA_TN8G"] = "1" if arm == "32" else arm
        env["MXA_TN8G32"] = "1" if arm == "32" else "0"
        env["ARM32"] = arm
        r = subprocess.run([sys.executable, "-c", CHILD], env=env,
                           capture_output=True, text=True,
                           cwd=os.path.dirname(os.path.dirname(
                               os.path.abspath(__file__))))
        print(r.stdout.strip())
        if r.returncode != 0:
            print("ARM", arm, "FAILED:", r.stderr[-2000:])


if __name__ == "__main__":
    main()
