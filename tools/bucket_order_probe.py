"""Does ascending-bucket (page-sequential) access speed the sparse update?
Compares store_update on keys whose buckets are random (hash low bits of
random u64) vs keys crafted so buckets ascend with the same total spread."""
import os
import sys
import time

sys.path.insert(0, os.path.join(os.path.dirname(__file__), ".."))

import numpy as np
import torch

from persia_amd.core.store import HipEmbeddingStore
from persia_amd.embedding import EmbeddingConfig
from persia_amd.embedding.optim import Adagrad

dev = torch.device("cuda", 0)
torch.cuda.set_device(dev)
U, dim, cap = 212992, 128, 100_000_000


def run(label, keys_np):
    store = HipEmbeddingStore(dim, cap, Adagrad(lr=0.1), EmbeddingConfig(), dev)
    keys = torch.from_numpy(keys_np.view(np.int64)).to(dev)
    store.lookup(keys, train=True)  # insert
    grads = torch.randn(U, dim, device=dev) * 0.01
    for _ in range(5):
        store.update_gradients(keys, grads)
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(30):
        store.update_gradients(keys, grads)
    torch.cuda.synchronize()
    us = (time.perf_counter() - t0) / 30 * 1e6
    gb = U * (dim * 2 * 4 * 2 + dim * 4) / 1e9  # row rd+wr + grad rd
    print(f"{label}: {us:7.1f} us  {gb/(us/1e6):5.2f} GB -> {gb/us*1e3:5.2f} TB/s",
          flush=True)
    del store
    torch.cuda.empty_cache()


rng = np.random.default_rng(0)
# random buckets: uniform random mixed keys (the production pattern)
rand_keys = rng.integers(1, 2**63, size=U, dtype=np.uint64)
# ascending buckets: same count, buckets stride evenly through the table
nb = (1 << 23)  # n_buckets for cap 1e8 (floor_pow2(cap/8))
stride = nb // U if nb > U else 1
asc_buckets = (np.arange(U, dtype=np.uint64) * np.uint64(max(1, stride)))
asc_keys = asc_buckets | (rng.integers(1, 2**30, size=U, dtype=np.uint64) << np.uint64(32))
run("random-buckets ", np.sort(rand_keys))
run("ascending-bucket", asc_keys)
