"""Why do flat-view params/grads slow hipBLASLt-path models? Time eager
fwd+bwd of an MLP with (none|preset|flat-view) .grad handling."""
import time

import os
import sys

sys.path.insert(0, os.path.join(os.path.dirname(__file__), ".."))

import torch

dev = "cuda"
torch.manual_seed(0)


def bench(mode):
    m = torch.nn.Sequential(
        *[torch.nn.Linear(1024, 1024, device=dev).bfloat16() for _ in range(6)]
    )
    ps = list(m.parameters())
    if mode == "flat":
        n = sum(p.numel() for p in ps)
        fw = torch.zeros(n, dtype=torch.bfloat16, device=dev)
        fg = torch.zeros_like(fw)
        off = 0
        for p in ps:
            k = p.numel()
            fw[off:off + k].copy_(p.detach().view(-1))
            with torch.no_grad():
                p.data = fw[off:off + k].view_as(p)
            p.grad = fg[off:off + k].view_as(p)
            off += k
    elif mode == "flatw_only":  # flat weights, standalone grads
        n = sum(p.numel() for p in ps)
        fw = torch.zeros(n, dtype=torch.bfloat16, device=dev)
        off = 0
        for p in ps:
            k = p.numel()
            fw[off:off + k].copy_(p.detach().view(-1))
            with torch.no_grad():
                p.data = fw[off:off + k].view_as(p)
            off += k
        for p in ps:
            p.grad = torch.zeros_like(p)
    elif mode == "preset":
        for p in ps:
            p.grad = torch.zeros_like(p)
    x = torch.randn(8192, 1024, device=dev, dtype=torch.bfloat16)
    for _ in range(5):
        m(x).sum().backward()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(20):
        m(x).sum().backward()
    torch.cuda.synchronize()
    g = ps[0].grad
    print(mode, round((time.perf_counter() - t0) / 20 * 1e3, 3), "ms",
          "grad_is_view:", g._base is not None if g is not None else None,
          flush=True)


for mode in ("none", "preset", "flatw_only", "flat"):
    bench(mode)
