import torch, sys
sys.path.insert(0, ".")
from anand_mxnet_amd import _C
M = N = K = 128
# A[r, k] = delta(k == kprobe), B[n, k] = n + ... probe which k-bytes a lane reads
for kprobe in [0, 1, 8, 16, 63, 64, 127]:
    a = torch.zeros((M, K), dtype=torch.int8, device="cuda")
    a[:, kprobe] = 1
    b = torch.arange(N, dtype=torch.int8, device="cuda").unsqueeze(1).expand(N, K).contiguous()
    # ref: C[r, n] = b[n, kprobe] = n
    out = _C.gemm_i8(a, b)
    ref = (a.float() @ b.float().t()).round().to(torch.int32)
    ok = (out == ref).all().item()
    if not ok:
        bad = (out != ref)
        ij = bad.nonzero()[:4].tolist()
        print(f"kprobe={kprobe}: MISMATCH at {ij}; out={[out[i, j].item() for i, j in ij]} ref={[ref[i, j].item() for i, j in ij]}")
    else:
        print(f"kprobe={kprobe}: ok")
# random small sanity
torch.manual_seed(0)
a = torch.randint(-4, 5, (M, K), dtype=torch.int8, device="cuda")
b = torch.randint(-4, 5, (N, K), dtype=torch.int8, device="cuda")
out = _C.gemm_i8(a, b); ref = (a.float() @ b.float().t()).round().to(torch.int32)
d = (out - ref)
print("rand max|diff|:", d.abs().max().item(), "frac wrong:", (d != 0).float().mean().item())
