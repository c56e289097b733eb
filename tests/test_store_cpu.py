import numpy as np
import torch

from persia_amd.core import hashing
from persia_amd.core.store import BUCKET_SIZE, PROBE_BUCKETS, CpuEmbeddingStore, row_init
from persia_amd.embedding import EmbeddingConfig
from persia_amd.embedding.optim import SGD, Adagrad, Adam


def _keys(signs):
    h = hashing.splitmix64(np.asarray(signs, dtype=np.uint64))
    return torch.from_numpy(h.view(np.int64))


def _store(opt=None, capacity=4096, dim=4):
    return CpuEmbeddingStore(
        dim, capacity, opt or SGD(lr=0.1), EmbeddingConfig(emb_initialization=(-0.5, 0.5))
    )


def test_insert_and_deterministic_init():
    s = _store()
    k = _keys([1, 2, 3])
    rows = s.lookup(k, train=True)
    assert rows.shape == (3, 4)
    assert len(s) == 3
    # same sign -> same init regardless of lookup order / store instance
    s2 = _store()
    rows2 = s2.lookup(_keys([3, 1, 2]), train=True)
    assert torch.allclose(rows[0], rows2[1])
    assert torch.allclose(rows[2], rows2[0])
    # matches the documented row_init function
    expected = torch.from_numpy(row_init(1, 4, -0.5, 0.5))
    assert torch.allclose(rows[0], expected)
    # bounded
    assert rows.abs().max() <= 0.5


def test_infer_miss_returns_zeros():
    s = _store()
    s.lookup(_keys([1]), train=True)
    rows = s.lookup(_keys([1, 99]), train=False)
    assert not torch.all(rows[0] == 0)
    assert torch.all(rows[1] == 0)
    assert len(s) == 1  # no insert in infer mode


def test_update_gradients_sgd():
    s = _store()
    k = _keys([7])
    before = s.lookup(k, train=True).clone()
    g = torch.ones(1, 4)
    skipped = s.update_gradients(k, g)
    assert skipped == 0
    after = s.lookup(k, train=True)
    assert torch.allclose(after, before - 0.1 * g, atol=1e-6)
    # missing sign is skipped, not crashed
    assert s.update_gradients(_keys([12345]), g) == 1


def test_update_gradients_adagrad_state():
    opt = Adagrad(lr=0.1, initial_accumulator_value=0.01, g_square_momentum=1.0, eps=1e-10)
    s = _store(opt)
    k = _keys([7])
    w0 = s.lookup(k, train=True).clone()
    g = torch.full((1, 4), 2.0)
    s.update_gradients(k, g)
    w1 = s.lookup(k, train=True)
    assert torch.allclose(w1, w0 - 0.1 * 2.0 * torch.rsqrt(torch.tensor(0.01 + 1e-10)), atol=1e-5)
    # second update sees acc = 0.01 + 4
    s.update_gradients(k, g)
    w2 = s.lookup(k, train=True)
    assert torch.allclose(w2, w1 - 0.1 * 2.0 * torch.rsqrt(torch.tensor(4.01 + 1e-10)), atol=1e-5)


def test_adam_beta_powers_step():
    opt = Adam(lr=0.01)
    s = _store(opt)
    k = _keys([1])
    s.lookup(k, train=True)
    assert s.beta1_power == opt.betas[0]
    g = torch.ones(1, 4)
    s.update_gradients(k, g)
    assert abs(s.beta1_power - opt.betas[0] ** 2) < 1e-12
    s.update_gradients(k, g)
    assert abs(s.beta1_power - opt.betas[0] ** 3) < 1e-12


def test_eviction_bounded_capacity():
    # tiny store: one bucket's probe window = whole table
    s = CpuEmbeddingStore(2, BUCKET_SIZE * PROBE_BUCKETS, SGD(lr=0.1), EmbeddingConfig())
    n_slots = s.n_slots
    signs = np.arange(1, 10 * n_slots, dtype=np.uint64)
    for sign in signs:
        s.lookup(_keys([sign]), train=True)
    assert len(s) <= n_slots
    # recently used signs survive (approximate LRU): the last one must be there
    rows = s.lookup(_keys([signs[-1]]), train=False)
    assert not torch.all(rows == 0)


def test_export_import_roundtrip():
    s = _store(Adagrad(lr=0.1))
    k = _keys([5, 6, 7])
    s.lookup(k, train=True)
    s.update_gradients(k, torch.randn(3, 4))
    signs, inner = s.export_rows()
    assert sorted(signs.tolist()) == [5, 6, 7]
    s2 = _store(Adagrad(lr=0.1))
    s2.import_rows(signs, inner)
    r1 = s.lookup(k, train=False)
    r2 = s2.lookup(k, train=False)
    assert torch.allclose(r1, r2)
    # optimizer state also carried (row_width includes accumulator)
    assert inner.shape[1] == 8
