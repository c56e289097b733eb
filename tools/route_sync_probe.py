"""Find which op in engine._a2a_route blocks the host: run each op after
building a deep stream backlog; a synchronizing op's host time jumps to the
backlog length.  (PA route phase measured 2.27 ms/batch on the GPU.)"""
import time

import os
import sys

sys.path.insert(0, os.path.join(os.path.dirname(__file__), ".."))

import torch

dev = torch.device("cuda", 0)
torch.cuda.set_device(dev)
nnz = 212992
world, cap = 1, 212992
uniq = torch.randint(1, 2**60, (nnz,), dtype=torch.int64, device=dev)
u_count = torch.tensor([nnz - 5000], device=dev)
ar = torch.arange(nnz, device=dev)
ones = torch.ones(nnz, dtype=torch.int64, device=dev)
big = torch.randn(8192, 8192, device=dev)

torch.cuda.set_sync_debug_mode("warn")
for it in range(3):
    torch.cuda.synchronize()
    for _ in range(30):
        big = big @ big * 1e-3  # ~30 x ~1ms backlog
    t = {}

    def tick(name, fn):
        t0 = time.perf_counter()
        r = fn()
        t[name] = (time.perf_counter() - t0) * 1e3
        return r

    owner = tick("owner", lambda: ((uniq >> 32) & 0xFFFFFFFF) * world >> 32)
    owner = tick("where_valid", lambda: torch.where(
        ar < u_count, owner, torch.full_like(owner, world)))
    counts = torch.zeros(world + 1, dtype=torch.int64, device=dev)
    tick("scatter_add", lambda: counts.scatter_add_(0, owner, ones))
    starts = tick("cumsum", lambda: torch.cumsum(counts, 0) - counts)
    pos = tick("gather", lambda: ar - starts.gather(0, owner))
    bad = tick("bad", lambda: (owner >= world) | (pos >= cap))
    idx = tick("where_idx", lambda: torch.where(
        bad, torch.full_like(pos, world * cap), owner * cap + pos))
    send = tick("zeros", lambda: torch.zeros(
        world * cap + 1, dtype=torch.int64, device=dev))
    tick("scatter", lambda: send.scatter_(0, idx, uniq))
    ovf = torch.zeros(1, dtype=torch.int64, device=dev)
    tick("ovf", lambda: ovf.add_((counts[:world] > cap).sum()))
    tick("clone", lambda: send[: world * cap].clone())
    print({k: round(v, 3) for k, v in t.items()}, flush=True)
torch.cuda.synchronize()
print("done")
