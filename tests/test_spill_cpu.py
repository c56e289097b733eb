"""Host-DRAM spill tier semantics (CPU model; the HIP path is covered by
tests/test_gpu_kernels.py::test_spill_roundtrip_gpu)."""
import numpy as np
import pytest
import torch

from persia_amd.core import hashing
from persia_amd.core.store import BUCKET_SIZE, PROBE_BUCKETS, CpuEmbeddingStore
from persia_amd.embedding import EmbeddingConfig
from persia_amd.embedding.optim import Adagrad


def _keys(signs):
    h = hashing.splitmix64(np.asarray(signs, dtype=np.uint64))
    return torch.from_numpy(h.view(np.int64))


def test_evicted_row_survives_in_host_tier():
    store = CpuEmbeddingStore(
        4,
        BUCKET_SIZE * PROBE_BUCKETS,  # one probe window = whole table
        Adagrad(lr=0.1, initial_accumulator_value=0.01),
        EmbeddingConfig(emb_initialization=(-0.5, 0.5)),
        spill_capacity=10_000,
    )
    k7 = _keys([7])
    before = store.lookup(k7, train=True).clone()
    store.update_gradients(k7, torch.full((1, 4), 2.0))
    updated = store.lookup(k7, train=False).clone()
    assert not torch.equal(before, updated)

    # flood the table with other signs until sign 7 gets evicted
    sign = 1000
    while store._probe(np.uint64(int(hashing.splitmix64(np.array([7], np.uint64))[0]))) >= 0:
        store.lookup(_keys(list(range(sign, sign + 8))), train=True)
        sign += 8
        assert sign < 100000, "sign 7 never evicted?"
    assert len(store.spill) > 0

    # next training lookup restores the UPDATED row (incl. optimizer state)
    restored = store.lookup(k7, train=True)
    assert torch.equal(restored, updated)
    # and the accumulator survived: identical second update on both paths
    store2 = CpuEmbeddingStore(
        4, 1 << 12, Adagrad(lr=0.1, initial_accumulator_value=0.01),
        EmbeddingConfig(emb_initialization=(-0.5, 0.5)),
    )
    store2.lookup(k7, train=True)
    store2.update_gradients(k7, torch.full((1, 4), 2.0))
    store.update_gradients(k7, torch.full((1, 4), 0.5))
    store2.update_gradients(k7, torch.full((1, 4), 0.5))
    assert torch.allclose(
        store.lookup(k7, train=False), store2.lookup(k7, train=False), atol=1e-6
    )


def test_export_includes_spilled_rows():
    store = CpuEmbeddingStore(
        2, BUCKET_SIZE * PROBE_BUCKETS, Adagrad(lr=0.1), EmbeddingConfig(),
        spill_capacity=1 << 14,
    )
    for s in range(0, 400, 8):
        store.lookup(_keys(list(range(s, s + 8))), train=True)
    assert len(store.spill) > 0
    signs, inner = store.export_rows()
    assert len(signs) == len(store) + len(store.spill)
    assert inner.shape[1] == store.row_width


def test_native_host_tier_matches_python_oracle():
    """C++ NativeHostTier (csrc/engine.cpp) vs the python HostTier under a
    random insert/fetch/evict workload: identical membership, rows, sizes,
    and export contents."""
    pytest.importorskip("persia_amd._C")
    from persia_amd.core.store import HostTier, NativeHostTier

    rng = np.random.default_rng(0)
    py, nat = HostTier(50, 4), NativeHostTier(50, 4)
    for step in range(300):
        k = rng.integers(0, 120, size=rng.integers(1, 9), dtype=np.uint64)
        if step % 3 == 2:
            rp, fp = py.fetch(k)
            rn, fn = nat.fetch(k)
            assert np.array_equal(fp, fn), step
            assert np.allclose(rp, rn), step
        else:
            r = rng.normal(size=(len(k), 4)).astype(np.float32)
            py.insert(k, r)
            nat.insert(k, r)
        assert len(py) == len(nat), step
    kp, rp = py.export()
    kn, rn = nat.export()
    assert np.array_equal(np.sort(kp), np.sort(kn))
    assert np.allclose(rp[np.argsort(kp)], rn[np.argsort(kn)])


def test_spill_capacity_bounded():
    store = CpuEmbeddingStore(
        2, BUCKET_SIZE * PROBE_BUCKETS, Adagrad(lr=0.1), EmbeddingConfig(),
        spill_capacity=16,
    )
    for s in range(0, 2000, 8):
        store.lookup(_keys(list(range(s, s + 8))), train=True)
    assert len(store.spill) <= 16


def _mk_spill_engine(ctx):
    from persia_amd.core.engine import EmbeddingEngine
    from persia_amd.core.schema import EmbeddingSchema, GlobalConfig, SlotConfig

    # tiny capacity forces evictions into the spill tier
    return EmbeddingEngine(
        schema=EmbeddingSchema(slots={"a": SlotConfig(name="a", dim=8)}),
        hyper=EmbeddingConfig(emb_initialization=(-0.5, 0.5)),
        optimizer=Adagrad(lr=0.1),
        gconf=GlobalConfig(capacity=64, spill_capacity=4096),
        device=torch.device("cpu"),
        dist_ctx=ctx,
        wire_dtype=torch.float32,
    )


def _spill_batches():
    from persia_amd.embedding.data import IDTypeFeature, Label, PersiaBatch

    rng = np.random.default_rng(7)
    out = []
    for _step in range(4):
        feats = [IDTypeFeature("a", [
            rng.integers(0, 2000, size=4, dtype=np.uint64) for _ in range(8)
        ])]
        out.append(PersiaBatch(
            feats, labels=[Label(np.ones((8, 1), np.float32))],
            requires_grad=True,
        ))
    return out


def _spill_worker(rank, port, result_dir):
    import os

    import torch.distributed as dist

    from persia_amd.core.comm import DistContext

    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    dist.init_process_group("gloo", rank=rank, world_size=2)
    eng = _mk_spill_engine(DistContext.new_sparse_group())
    outs = [
        [p.sum_tensor.clone() for p in eng.process_batch(b).payloads]
        for b in _spill_batches()
    ]
    if rank == 0:
        import os.path

        torch.save(outs, os.path.join(result_dir, "spill_w2.pt"))
    dist.barrier()
    dist.destroy_process_group()


def test_spill_with_world2_generic_path(tmp_path):
    """Spill + distributed falls back to the exact two-phase exchange (the
    padded fast path requires spill=None): a 2-rank run with a spill tier
    must still produce the world-1 lookup results bitwise (f32 wire)."""
    import torch.multiprocessing as mp

    from persia_amd.core.comm import DistContext
    from persia_amd.utils import find_free_port

    ref_eng = _mk_spill_engine(DistContext(1, 0))
    ref = [
        [p.sum_tensor.clone() for p in ref_eng.process_batch(b).payloads]
        for b in _spill_batches()
    ]
    assert any(len(s.spill) > 0 for s in ref_eng.stores.values()), (
        "test setup must actually spill"
    )
    port = find_free_port()
    mp.spawn(_spill_worker, args=(port, str(tmp_path)), nprocs=2, join=True)
    got = torch.load(tmp_path / "spill_w2.pt")
    for step, (r, g) in enumerate(zip(ref, got)):
        for pr, pg in zip(r, g):
            assert torch.equal(pr, pg), f"step {step}"
