"""Per-shape tn8(bf16) vs mx8(fp8) on the transformer-big GEMM shapes."""
import torch, time, sys
sys.path.insert(0, ".")
from anand_mxnet_amd import _C
BF = torch.bfloat16
def t(f, n=30):
    for _ in range(3): f()
    torch.cuda.synchronize(); t0=time.perf_counter()
    for _ in range(n): f()
    torch.cuda.synchronize(); return (time.perf_counter()-t0)/n*1e3
for (M,N,K) in [(8192,1024,1024),(8192,4096,1024),(8192,1024,4096),
                (8192,3072,1024)]:
    a = (torch.rand((M,K), device="cuda")*2-1).to(BF)
    b = (torch.rand((N,K), device="cuda")*2-1).to(BF)
    ms_bf = t(lambda: _C.gemm_tn(a, b, None, False))
    a8, asc = _C.mx_cast(a)
    b8, bsc = _C.mx_cast(b)
    ms_mx = t(lambda: _C.gemm_mx(a8, asc, b8, bsc))
    ms_cast = t(lambda: _C.mx_cast(a))
    fl = 2.0*M*N*K/1e12
    print(f"{M}x{N}x{K}: bf16 {ms_bf:.3f}ms {fl/ms_bf*1e3:.0f}TF | "
          f"mx8 {ms_mx:.3f}ms {fl/ms_mx*1e3:.0f}TF | cast(a) {ms_cast:.3f}ms")
