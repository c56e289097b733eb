"""Wire-format micro-benchmarks: the MI355X-native analog of the
reference's criterion benches (others/persia-common-benchmark/benches/
serialize_inf_request.rs and benches/memcpy.rs).

* batch serialization: PersiaBatch.to_bytes / from_bytes round-trip
  throughput at an inference-request shape (the reference compares speedy
  vs serde codecs on the same payload; our wire is the documented
  numpy-backed binary layout in embedding/data.py);
* memcpy: host bytes/s at the batch size, plus pinned H2D when a GPU is
  visible (the reference measures raw memcpy as its transport floor).

Run: python tools/wire_bench.py  (CPU-only safe; GPU part auto-skips)
"""
import sys
import time

sys.path.insert(0, "/root/repo")
import numpy as np

from persia_amd.embedding.data import (
    IDTypeFeatureWithSingleID,
    Label,
    NonIDTypeFeature,
    PersiaBatch,
)


def make_batch(B=8192, n_slots=26, n_dense=13):
    rng = np.random.default_rng(0)
    feats = [
        IDTypeFeatureWithSingleID(
            f"f{i}", rng.integers(0, 1 << 40, size=B, dtype=np.uint64)
        )
        for i in range(n_slots)
    ]
    return PersiaBatch(
        feats,
        non_id_type_features=[
            NonIDTypeFeature(rng.random((B, n_dense), dtype=np.float32))
        ],
        labels=[Label(rng.random((B, 1), dtype=np.float32))],
        requires_grad=False,
    )


def bench(fn, n=50):
    fn()  # warm
    t0 = time.perf_counter()
    for _ in range(n):
        fn()
    return (time.perf_counter() - t0) / n


def main():
    batch = make_batch()
    buf = batch.to_bytes()
    mb = len(buf) / 1e6
    t_ser = bench(batch.to_bytes)
    t_de = bench(lambda: PersiaBatch.from_bytes(buf))
    print(f"batch wire size: {mb:.2f} MB  (B=8192, 26 id slots, 13 dense)")
    print(f"serialize:   {t_ser*1e3:7.3f} ms  {mb/1e3/t_ser:8.2f} GB/s")
    print(f"deserialize: {t_de*1e3:7.3f} ms  {mb/1e3/t_de:8.2f} GB/s")

    src = np.frombuffer(buf, dtype=np.uint8).copy()
    dst = np.empty_like(src)
    t_cp = bench(lambda: np.copyto(dst, src))
    print(f"host memcpy: {t_cp*1e3:7.3f} ms  {mb/1e3/t_cp:8.2f} GB/s")

    try:
        import torch

        if torch.cuda.is_available():
            pin = torch.from_numpy(src).pin_memory()
            dev = torch.empty_like(pin, device="cuda")

            def h2d():
                dev.copy_(pin, non_blocking=True)
                torch.cuda.synchronize()

            t_h2d = bench(h2d)
            print(f"pinned H2D:  {t_h2d*1e3:7.3f} ms  {mb/1e3/t_h2d:8.2f} GB/s")
    except ImportError:
        pass


if __name__ == "__main__":
    main()
