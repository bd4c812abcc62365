#!/usr/bin/env python3
"""Routing sweep: transformer-shape GEMMs on the 128^2 kernel (tn8 gated
off) vs the 2-barrier-window 256^2 kernel (gates lowered).  Decides the
MXA_TN8_MINK / MXA_TN8_MINBLOCKS defaults."""
import json
import os
import subprocess
import sys

# (M, N, K): bert_base / transformer_big fwd+wgrad-shaped dense GEMMs
SHAPES = [(16384, 3072, 768), (16384, 768, 3072), (16384, 768, 768),
          (16384, 4096, 1024), (16384, 1024, 4096), (32768, 3072, 768),
          (32768, 768, 768)]

CHILD = r"""
import json, os, sys, time
import torch
sys.path.insert(0, ".")
from anand_mxnet_amd import _C
BF = torch.bfloat16
DEV = torch.device("cuda", 0)
res = []
for (M, N, K) in json.loads(os.environ["AB_SHAPES"]):
    a = (torch.rand((M, K), device=DEV) * 2 - 1).to(BF)
    b = (torch.rand((N, K), device=DEV) * 2 - 1).to(BF)
    for _ in range(5):
        _C.gemm_tn(a, b, None, False)
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(30):
        _C.gemm_tn(a, b, None, False)
    torch.cuda.synchronize()
    t = (time.perf_counter() - t0) / 30
    res.append({"shape": f"{M}x{N}x{K}",
                "TF": round(2.0 * M * N * K / t / 1e12, 1)})
print(json.dumps({"arm": os.environ.get("ARM"), "gemms": res}))
"""


def main():
    env = dict(os.environ)
    env["AB_SHAPES"] = json.dumps(SHAPES)
    arms = [("128sq", {"MXA_TN8_MINBLOCKS": "99999"}),
            ("tn8g-lowgate", {"MXA_TN8_MINK": "256",
                              "MXA_TN8_MINBLOCKS": "64"})]
    for name, overrides in arms:
        e = dict(env)
        e.update(overrides)
        e["ARM"] = name
        r = subprocess.run([sys.executable, "-c", CHILD], env=e,
                           capture_output=True, text=True,
                           cwd=os.path.dirname(os.path.dirname(
                               os.path.abspath(__file__))))
        print(r.stdout.strip())
        if r.returncode != 0:
            print("ARM", name, "FAILED:", r.stderr[-1500:])


if __name__ == "__main__":
    main()
