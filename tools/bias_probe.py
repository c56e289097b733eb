"""Micro-benchmark bias_grad variants vs torch reference sum(0)."""
import time

import os
import sys

sys.path.insert(0, os.path.join(os.path.dirname(__file__), ".."))

import torch

from persia_amd.ops import native

C = native()
dev = torch.device("cuda", 0)
for M, N in ((8192, 512), (8192, 256), (8192, 128), (8192, 1024)):
    dC = torch.randn(M, N, device=dev).to(torch.bfloat16)
    ref = dC.float().sum(0)
    db = C.bias_grad(dC)
    assert torch.allclose(db, ref, atol=3.0, rtol=0.02), (M, N, (db-ref).abs().max())
    for name, fn in (("ours", lambda: C.bias_grad(dC)),
                     ("torch", lambda: dC.float().sum(0)),
                     ("torch_bf", lambda: dC.sum(0))):
        for _ in range(10):
            fn()
        torch.cuda.synchronize()
        t0 = time.perf_counter()
        for _ in range(50):
            fn()
        torch.cuda.synchronize()
        print(f"{M}x{N} {name}: {(time.perf_counter()-t0)/50*1e6:.1f} us", flush=True)
