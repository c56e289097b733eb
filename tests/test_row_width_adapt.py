"""Checkpoint rows adapt across optimizer-state widths at import (reference
infer PS slices the emb prefix on lookup, mod.rs:231-251; we adapt the row
at import to keep the fixed arena layout — core/store.py _adapt_row_width):

* Adagrad train dump (dim emb + dim accumulator) -> SGD/infer store
  (no state): state columns truncated, embeddings preserved bit-exact;
* SGD dump -> Adagrad store: state columns re-initialized, training
  continues (accumulator starts at the optimizer's init value).

Stores address rows by MIXED keys (splitmix64 of the sign) while
export/import speak the sign space — the tests mix explicitly like the
engine's sign_prep does.
"""
import numpy as np
import pytest
import torch

from persia_amd.core import hashing


def _store(optim, capacity=1 << 10, dim=8):
    from persia_amd.core.store import CpuEmbeddingStore
    from persia_amd.embedding import EmbeddingConfig

    return CpuEmbeddingStore(
        dim=dim, capacity=capacity, optimizer=optim,
        hyper=EmbeddingConfig(), device=torch.device("cpu"),
    )


def _mix(signs: np.ndarray) -> torch.Tensor:
    return torch.from_numpy(
        hashing.splitmix64(signs.astype(np.uint64)).view(np.int64)
    )


def test_adagrad_dump_to_infer_sgd_store():
    from persia_amd.embedding.optim import SGD, Adagrad

    train = _store(Adagrad(lr=0.1))
    signs0 = np.arange(1, 65, dtype=np.uint64)
    train.lookup(_mix(signs0), train=True)
    train.update_gradients(_mix(signs0), torch.full((64, 8), 0.25))
    signs, inner = train.export_rows()
    assert inner.shape[1] == 16  # dim + accumulator
    assert sorted(signs) == sorted(signs0)

    infer = _store(SGD(lr=0.1))
    assert infer.row_width == 8
    infer.import_rows(signs, inner)
    # embedding prefix preserved bit-exact, state dropped
    got = infer.lookup(_mix(signs), train=False)
    want = train.lookup(_mix(signs), train=False)
    assert got.abs().sum() > 0, "imported rows not found (vacuous compare)"
    assert torch.equal(got, want[:, :8])


def test_sgd_dump_to_adagrad_store_reinits_state():
    from persia_amd.embedding.optim import SGD, Adagrad

    src = _store(SGD(lr=0.1))
    signs0 = np.arange(100, 132, dtype=np.uint64)
    src.lookup(_mix(signs0), train=True)
    signs, inner = src.export_rows()
    assert inner.shape[1] == 8

    dst = _store(Adagrad(lr=0.1, initial_accumulator_value=0.01))
    dst.import_rows(signs, inner)
    s2, inner2 = dst.export_rows()
    assert inner2.shape[1] == 16
    # embeddings preserved; accumulator columns at the optimizer init
    lut = {int(k): i for i, k in enumerate(s2)}
    for i, k in enumerate(signs):
        j = lut[int(k)]
        np.testing.assert_array_equal(inner2[j, :8], inner[i])
        assert np.allclose(inner2[j, 8:], dst.optimizer.initial_accumulator_value)
    # and an update step works on the imported rows (none skipped)
    n = dst.update_gradients(_mix(signs), torch.full((32, 8), 0.5))
    assert n == 0
