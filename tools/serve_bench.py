"""Serving-path latency benchmark: byte-wire request -> infer lookup ->
dense forward, at a criteo-shaped request (26 id slots, 13 dense, dim 128).

The reference serves through TorchServe+gRPC and quotes no latency numbers;
this measures OUR end-to-end handler path (deserialize + read-only HBM
lookup + bf16 MFMA forward) per request batch on one MI355X, p50/p99 over
wall-clock per request.

Run: python tools/serve_bench.py [batch_size ...]
"""
import sys
import time

sys.path.insert(0, "/root/repo")
import numpy as np
import torch

from persia_amd.core.comm import DistContext
from persia_amd.core.engine import EmbeddingEngine
from persia_amd.core.schema import EmbeddingSchema, GlobalConfig, SlotConfig
from persia_amd.embedding import EmbeddingConfig
from persia_amd.embedding.data import (
    IDTypeFeatureWithSingleID,
    NonIDTypeFeature,
    PersiaBatch,
)
from persia_amd.embedding.optim import Adagrad
from persia_amd.models import DLRM

N_SLOTS, N_DENSE, DIM = 26, 13, 128


def make_request(B, rng):
    feats = [
        IDTypeFeatureWithSingleID(
            f"f{i}", rng.integers(0, 1 << 27, size=B, dtype=np.uint64)
        )
        for i in range(N_SLOTS)
    ]
    return PersiaBatch(
        feats,
        non_id_type_features=[
            NonIDTypeFeature(rng.random((B, N_DENSE), dtype=np.float32))
        ],
        requires_grad=False,
    ).to_bytes()


def main():
    sizes = [int(a) for a in sys.argv[1:]] or [1, 64, 1024]
    use_gpu = torch.cuda.is_available()
    device = torch.device("cuda", 0) if use_gpu else torch.device("cpu")
    rng = np.random.default_rng(0)
    schema = EmbeddingSchema(
        slots={f"f{i}": SlotConfig(name=f"f{i}", dim=DIM)
               for i in range(N_SLOTS)}
    )
    engine = EmbeddingEngine(
        schema=schema, hyper=EmbeddingConfig(), optimizer=Adagrad(lr=0.01),
        gconf=GlobalConfig(capacity=1 << 24, job_type="infer"),
        device=device, dist_ctx=DistContext(1, 0),
    )
    model = DLRM(num_sparse=N_SLOTS, num_dense=N_DENSE, dim=DIM,
                 fused=use_gpu).to(device)
    if use_gpu:
        model = model.bfloat16()
        # FusedLinear keeps biases f32 through .bfloat16(); dense input cast
        # happens in the loop
    model.eval()

    # warm the table so lookups hit resident rows (serving steady state)
    warm = PersiaBatch(
        [IDTypeFeatureWithSingleID(
            f"f{i}", rng.integers(0, 1 << 27, size=4096, dtype=np.uint64))
         for i in range(N_SLOTS)],
        non_id_type_features=[
            NonIDTypeFeature(rng.random((4096, N_DENSE), dtype=np.float32))
        ],
        requires_grad=False,
    )
    engine.process_batch(warm, train=True)

    for B in sizes:
        reqs = [make_request(B, rng) for _ in range(32)]
        lat = []

        def serve(buf):
            tb = engine.process_batch(PersiaBatch.from_bytes(buf), train=False)
            embs = [p.sum_tensor for p in tb.payloads]
            dense = tb.non_id_type_tensors[0]
            if use_gpu:
                dense = dense.bfloat16()
                base = tb._groups[0].sum_base
                with torch.no_grad():
                    out = model(dense, base)
            else:
                with torch.no_grad():
                    out = model(tb.non_id_type_tensors, embs)
            if use_gpu:
                torch.cuda.synchronize()
            return out

        for buf in reqs[:8]:
            serve(buf)  # warm
        for buf in reqs:
            t0 = time.perf_counter()
            serve(buf)
            lat.append((time.perf_counter() - t0) * 1e3)
        lat.sort()
        p50 = lat[len(lat) // 2]
        p99 = lat[min(len(lat) - 1, int(len(lat) * 0.99))]
        qps = B / (sum(lat) / len(lat) / 1e3)
        print(f"B={B:5d}: p50 {p50:7.3f} ms  p99 {p99:7.3f} ms  "
              f"{qps/1e3:8.1f}k samples/s", flush=True)


if __name__ == "__main__":
    main()
