"""dgrad strategies: v1 trans_b staging vs (transpose W + NT path)."""
import os
import sys
import time

sys.path.insert(0, os.path.join(os.path.dirname(__file__), ".."))

import torch

from persia_amd.ops import native

C = native()
dev = torch.device("cuda", 0)
torch.manual_seed(0)


def bench(fn, n=50):
    for _ in range(8):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(n):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / n * 1e6


empty = torch.empty(0, device=dev)
for M, N, K in [(8192, 1024, 1024), (8192, 1024, 512), (8192, 512, 512),
                (8192, 256, 128), (8192, 512, 64)]:
    # dgrad: dX[M,K] = g[M,N] @ W[N,K]
    g = (torch.randn(M, N, device=dev) * 0.1).to(torch.bfloat16).contiguous()
    W = (torch.randn(N, K, device=dev) * 0.1).to(torch.bfloat16).contiguous()
    t_v1 = bench(lambda: C.gemm_nt_bias_act(g, W, empty, 0, 0, 1))
    Wt = W.t().contiguous()
    t_nt = bench(lambda: C.gemm_nt_bias_act(g, Wt, empty, 0, 0, 0))
    t_tr = bench(lambda: W.t().contiguous())
    ref = C.gemm_nt_bias_act(g, W, empty, 0, 0, 1)
    out = C.gemm_nt_bias_act(g, Wt, empty, 0, 0, 0)
    ok = torch.allclose(ref.float(), out.float(), atol=0.5, rtol=0.05)
    print(f"{M}x{N}x{K}: trans_b {t_v1:6.1f}us | NT(Wt) {t_nt:6.1f}us "
          f"+ tr {t_tr:4.1f}us | match={ok}", flush=True)
