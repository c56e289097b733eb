"""Correctness sweep + A/B of gemm_nt v2 (PA_GEMM_V2 chooses the build)."""
import os
import sys
import time

sys.path.insert(0, os.path.join(os.path.dirname(__file__), ".."))

import torch

from persia_amd.ops import native

C = native()
dev = torch.device("cuda", 0)
torch.manual_seed(0)
bad = 0
for M in (128, 192, 256, 1000, 8192):
    for N in (128, 256, 512, 1024):
        for K in (64, 448, 1728):
            A = (torch.randn(M, K, device=dev) * 0.3).to(torch.bfloat16)
            B = (torch.randn(N, K, device=dev) * 0.3).to(torch.bfloat16)
            bias = torch.randn(N, device=dev)
            out = C.gemm_nt_bias_act(A.contiguous(), B.contiguous(), bias, 0, 0, 0)
            ref = A.float() @ B.float().t() + bias
            err = (out.float() - ref).abs().max().item()
            tol = 0.1 + 0.02 * K ** 0.5
            if err > tol:
                bad += 1
                print(f"FAIL {M}x{N}x{K}: max err {err:.3f} tol {tol:.3f}", flush=True)
print("sweep done, failures:", bad, flush=True)


def bench(M, N, K, n=50):
    A = (torch.randn(M, K, device=dev) * 0.1).to(torch.bfloat16).contiguous()
    B = (torch.randn(N, K, device=dev) * 0.1).to(torch.bfloat16).contiguous()
    bias = torch.randn(N, device=dev)
    fn = lambda: C.gemm_nt_bias_act(A, B, bias, 1, 0, 0)
    for _ in range(8):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(n):
        fn()
    torch.cuda.synchronize()
    us = (time.perf_counter() - t0) / n * 1e6
    print(f"{M}x{N}x{K}: {us:6.1f} us {2*M*N*K/us/1e6:6.1f} TF", flush=True)


for shape in [(8192, 512, 512), (8192, 1024, 512), (8192, 1024, 1024),
              (8192, 512, 1024), (8192, 256, 512), (8192, 1024, 1728),
              (8192, 128, 64), (8192, 512, 64)]:
    bench(*shape)
