"""wgrad v2 (tr-read) correctness + perf vs v1."""
import os
import sys
import time

sys.path.insert(0, os.path.join(os.path.dirname(__file__), ".."))

import torch

from persia_amd.ops import native

C = native()
dev = torch.device("cuda", 0)
torch.manual_seed(0)
for (M, N, K) in [(128, 128, 128), (8192, 512, 512), (8192, 1024, 512),
                  (8192, 1024, 1024), (8192, 512, 128), (1024, 128, 384)]:
    dC = (torch.randn(M, N, device=dev) * 0.1).to(torch.bfloat16).contiguous()
    A = (torch.randn(M, K, device=dev) * 0.1).to(torch.bfloat16).contiguous()
    dW = C.wgrad(dC, A)
    ref = dC.float().t() @ A.float()
    err = (dW - ref).abs().max().item()
    ok = err < 0.5
    for _ in range(8):
        C.wgrad(dC, A)
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(50):
        C.wgrad(dC, A)
    torch.cuda.synchronize()
    us = (time.perf_counter() - t0) / 50 * 1e6
    print(f"M{M} N{N} K{K}: err={err:.4f} ok={ok} {us:7.1f}us "
          f"{2*M*N*K/us/1e6:6.1f}TF", flush=True)
