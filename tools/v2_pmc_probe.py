"""PMC probe of the FINAL round-2 kernels (glds GEMM v2 + tr-read wgrad)
at the bench's dominant padded shapes.  Run under:

  rocprofv3 --pmc SQ_WAVE_CYCLES SQ_WAIT_ANY SQ_WAIT_INST_ANY \
      SQ_ACTIVE_INST_ANY SQ_VALU_MFMA_BUSY_CYCLES SQ_INSTS_LDS \
      SQ_LDS_BANK_CONFLICT -d <out> -- python tools/v2_pmc_probe.py

Each shape runs in its own short loop; counter rows attribute per dispatch,
so the digest groups by (kernel, grid).
"""
import sys
import time

sys.path.insert(0, "/root/repo")
import torch

from persia_amd.ops import native

C = native()
dev = torch.device("cuda", 0)
torch.manual_seed(0)


def gemm(M, N, K, n=30):
    A = (torch.randn(M, K, device=dev) * 0.1).to(torch.bfloat16).contiguous()
    B = (torch.randn(N, K, device=dev) * 0.1).to(torch.bfloat16).contiguous()
    bias = torch.randn(N, device=dev)
    for _ in range(3):
        C.gemm_nt_bias_act(A, B, bias, 1, 0, 0)
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(n):
        C.gemm_nt_bias_act(A, B, bias, 1, 0, 0)
    torch.cuda.synchronize()
    dt = (time.perf_counter() - t0) / n
    print(f"gemm_nt {M}x{N}x{K}: {dt*1e6:.1f} us {2*M*N*K/dt/1e12:.1f} TF",
          flush=True)


def wgrad(M, N, K, n=30):
    dC = (torch.randn(M, N, device=dev) * 0.1).to(torch.bfloat16).contiguous()
    A = (torch.randn(M, K, device=dev) * 0.1).to(torch.bfloat16).contiguous()
    for _ in range(3):
        C.wgrad(dC, A)
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(n):
        C.wgrad(dC, A)
    torch.cuda.synchronize()
    dt = (time.perf_counter() - t0) / n
    print(f"wgrad {M}x{N}x{K}: {dt*1e6:.1f} us {2*M*N*K/dt/1e12:.1f} TF",
          flush=True)


gemm(8192, 1024, 1024)   # criteo top-MLP 1024x1024 layer
gemm(8192, 1024, 512)    # top-MLP first layer (interaction out, K padded)
wgrad(8192, 1024, 1024)  # its weight grad (reduction over the batch)
