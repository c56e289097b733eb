"""Timing/PMC probe: the big-layer wgrad shape, kernel launched in a loop so
counter runs attribute cleanly to wgrad_kernel."""
import sys, time, torch
sys.path.insert(0, "/root/repo")
from persia_amd.ops import native
C = native()
dev = torch.device("cuda", 0)
torch.manual_seed(0)
M, N, K = 8192, 1024, 1024
dC = (torch.randn(M, N, device=dev) * 0.1).to(torch.bfloat16).contiguous()
A = (torch.randn(M, K, device=dev) * 0.1).to(torch.bfloat16).contiguous()
for _ in range(5):
    C.wgrad(dC, A)
torch.cuda.synchronize()
t0 = time.perf_counter()
n = 50
for _ in range(n):
    C.wgrad(dC, A)
torch.cuda.synchronize()
dt = (time.perf_counter() - t0) / n
tf = 2 * M * N * K / dt / 1e12
print(f"wgrad {M}x{N}x{K}: {dt*1e6:.1f} us  {tf:.1f} TFLOP/s", flush=True)
