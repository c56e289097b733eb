import sys, time, torch
sys.path.insert(0, "/root/repo")
from persia_amd.ops import native
C = native()
dev = torch.device("cuda", 0)

def bench(fn, n=50):
    for _ in range(5): fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(n): fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / n * 1e6  # us

shapes = [(8192,512,13+19),(8192,256,512),(8192,128,256),(8192,1024,480),
          (8192,1024,1024),(8192,512,1024),(8192,256,512)]
print(f"{'M':>6}{'N':>6}{'K':>6} {'ours_us':>9} {'blaslt_us':>10} {'ours_TF':>9}")
for M,N,K in shapes:
    K = (K+31)//32*32
    A = (torch.randn(M,K,device=dev)*0.1).to(torch.bfloat16).contiguous()
    B = (torch.randn(N,K,device=dev)*0.1).to(torch.bfloat16).contiguous()
    bias = torch.randn(N, device=dev)
    t_ours = bench(lambda: C.gemm_nt_bias_act(A,B,bias,1,0,0))
    t_ref = bench(lambda: torch.relu(A @ B.t() + bias.to(torch.bfloat16)))
    tf = 2*M*N*K/t_ours/1e6
    print(f"{M:>6}{N:>6}{K:>6} {t_ours:9.1f} {t_ref:10.1f} {tf:9.1f}")

# wgrad
for M,N,K in [(8192,512,32),(8192,1024,1024),(8192,256,512)]:
    dC = (torch.randn(M,N,device=dev)*0.1).to(torch.bfloat16).contiguous()
    A = (torch.randn(M,K,device=dev)*0.1).to(torch.bfloat16).contiguous()
    t_ours = bench(lambda: C.wgrad(dC,A))
    t_ref = bench(lambda: dC.float().t() @ A.float())
    print(f"wgrad {M}x{N}x{K}: ours {t_ours:.1f}us ref {t_ref:.1f}us  {2*M*N*K/t_ours/1e6:.1f} TF")
