"""PMC probe of gemm_nt at the bench's dominant shape (what bounds it?)."""
import sys
import time

sys.path.insert(0, "/root/repo")
import torch

from persia_amd.ops import native

C = native()
dev = torch.device("cuda", 0)
torch.manual_seed(0)
M, N, K = 8192, 1024, 480
A = (torch.randn(M, K, device=dev) * 0.1).to(torch.bfloat16).contiguous()
B = (torch.randn(N, K, device=dev) * 0.1).to(torch.bfloat16).contiguous()
bias = torch.randn(N, device=dev)
for _ in range(5):
    C.gemm_nt_bias_act(A, B, bias, 1, 0, 0)
torch.cuda.synchronize()
t0 = time.perf_counter()
n = 50
for _ in range(n):
    C.gemm_nt_bias_act(A, B, bias, 1, 0, 0)
torch.cuda.synchronize()
dt = (time.perf_counter() - t0) / n
print(f"gemm_nt {M}x{N}x{K}: {dt*1e6:.1f} us {2*M*N*K/dt/1e12:.1f} TF", flush=True)
