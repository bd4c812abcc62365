import torch, time, sys
sys.path.insert(0, ".")
from anand_mxnet_amd import _C
BF = torch.bfloat16
def t(f, n=20):
    for _ in range(3): f()
    torch.cuda.synchronize(); t0=time.perf_counter()
    for _ in range(n): f()
    torch.cuda.synchronize(); return (time.perf_counter()-t0)/n*1e3
import os
for (N,H,C,K) in [(256,56,64,64),(256,28,128,128),(256,14,256,256),(256,7,512,512),(64,56,64,64),(64,28,128,128)]:
    x = (torch.rand((N,H,H,C), device="cuda")*2-1).to(BF)
    dy = (torch.rand((N,H,H,K), device="cuda")*2-1).to(BF)
    ms = t(lambda: _C.conv2d_wgrad(x, dy, [3,3],[1,1],[1,1],[1,1]))
    tf = 2.0*N*H*H*C*K*9/ (ms/1e3) / 1e12
    print(f"wgrad[{os.environ.get('MXA_WG_C9','1')}] b{N} {H}x{H}x{C}->K{K}: {ms:.3f} ms  {tf:.0f} TF")
