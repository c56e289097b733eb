"""PMC/timing probe of the packed interaction kernels at the criteo shape."""
import sys
import time

sys.path.insert(0, "/root/repo")
import torch

from persia_amd.ops import native

C = native()
dev = torch.device("cuda", 0)
torch.manual_seed(0)
B, S, D = 8192, 26, 128
F = S + 1
n_inter = F * (F - 1) // 2
x = (torch.randn(B, D, device=dev) * 0.5).to(torch.bfloat16).contiguous()
base = (torch.randn(S * B, D, device=dev) * 0.5).to(torch.float16).contiguous()
g = (torch.randn(B, n_inter, device=dev) * 0.5).to(torch.bfloat16).contiguous()


def bench(fn, n=30):
    for _ in range(5):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(n):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / n * 1e6


t_f = bench(lambda: C.interact_fwd_packed(x, base))
t_b = bench(lambda: C.interact_bwd_packed(g, x, base))
flops_f = 2 * B * F * F * D
flops_b = 2 * flops_f
print(f"fwd {t_f:6.1f} us  {flops_f/t_f/1e6:6.1f} TF   "
      f"bwd {t_b:6.1f} us  {flops_b/t_b/1e6:6.1f} TF", flush=True)
