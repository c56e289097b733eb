"""Bisect hipGraph capture of the dense iteration (fwd / +bwd / +opt)."""
import sys

import torch

sys.path.insert(0, "/root/repo")
from persia_amd.models import DLRM

torch.cuda.set_device(0)
dev = torch.device("cuda", 0)
B, S, D = 4096, 26, 128
model = DLRM(num_sparse=S, num_dense=13, dim=D).to(dev)
opt = torch.optim.SGD(model.parameters(), lr=0.01)
loss_fn = torch.nn.functional.binary_cross_entropy_with_logits

dense = torch.zeros(B, 13, device=dev)
base = torch.zeros(S * B, D, dtype=torch.float16, device=dev, requires_grad=True)
base.grad = torch.zeros_like(base)
label = torch.zeros(B, device=dev)

mode = sys.argv[1] if len(sys.argv) > 1 else "full"


def iteration():
    if mode != "fwd":
        base.grad.zero_()
    with torch.autocast("cuda", dtype=torch.bfloat16, cache_enabled=False):
        logits = model(dense, base)
        loss = loss_fn(logits.float(), label)
    if mode != "fwd":
        loss.backward()
    if mode == "full":
        opt.step()
        for p in model.parameters():
            p.grad.zero_()
    return loss


s = torch.cuda.Stream()
s.wait_stream(torch.cuda.current_stream())
with torch.cuda.stream(s):
    for _ in range(3):
        iteration()
torch.cuda.current_stream().wait_stream(s)
torch.cuda.synchronize()
print("warmup ok", flush=True)

g = torch.cuda.CUDAGraph()
with torch.cuda.graph(g):
    loss = iteration()
torch.cuda.synchronize()
print("capture ok", flush=True)

# check grad aliasing: does backward still write the tensor we hold?
g0 = base.grad.data_ptr()
for i in range(20):
    g.replay()
torch.cuda.synchronize()
print("replay ok; base.grad ptr stable:", base.grad.data_ptr() == g0,
      "grad norm:", float(base.grad.float().norm()), flush=True)

import time
t0 = time.perf_counter()
for i in range(100):
    g.replay()
torch.cuda.synchronize()
print(f"replay: {(time.perf_counter()-t0)/100*1000:.3f} ms/iter", flush=True)
