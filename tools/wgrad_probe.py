import sys, torch
sys.path.insert(0, "/root/repo")
from persia_amd.ops import native
C = native()
dev = torch.device("cuda", 0)
torch.manual_seed(0)
for (M, N, K) in [(512,256,224),(512,256,256),(1024,256,224),(512,64,64),(512,256,32),(4096,256,224),(512,1024,224)]:
    dC = (torch.randn(M, N, device=dev) * 0.1).to(torch.bfloat16).contiguous()
    A = (torch.randn(M, K, device=dev) * 0.1).to(torch.bfloat16).contiguous()
    dW = C.wgrad(dC, A)
    ref = dC.float().t() @ A.float()
    err = (dW - ref).abs().max().item()
    tiles = ((N+63)//64)*((K+63)//64)
    splitm = 1
    while tiles * splitm < 512 and splitm < 64 and (M // (splitm*2)) >= 32:
        splitm *= 2
    print(f"M{M} N{N} K{K} tiles{tiles} splitm{splitm} err {err:.4f}", flush=True)
