import numpy as np
import torch

from persia_amd.core.comm import DistContext
from persia_amd.core.engine import EmbeddingEngine
from persia_amd.core.schema import EmbeddingSchema, GlobalConfig, SlotConfig
from persia_amd.core.store import row_init
from persia_amd.core import hashing
from persia_amd.embedding import EmbeddingConfig
from persia_amd.embedding.data import IDTypeFeature, Label, PersiaBatch
from persia_amd.embedding.optim import SGD


def _schema(sum_a=True, dim=4, sfs=3):
    return EmbeddingSchema(
        slots={
            "a": SlotConfig(name="a", dim=dim, embedding_summation=sum_a,
                            sample_fixed_size=sfs),
            "b": SlotConfig(name="b", dim=dim, embedding_summation=True),
        },
        feature_index_prefix_bit=8,
    )


def _engine(schema, **kw):
    return EmbeddingEngine(
        schema=schema,
        hyper=EmbeddingConfig(emb_initialization=(-0.5, 0.5)),
        optimizer=SGD(lr=0.1),
        gconf=GlobalConfig(capacity=1 << 14),
        device=torch.device("cpu"),
        dist_ctx=DistContext(1, 0),
        **kw,
    )


def _batch(requires_grad=True):
    a = IDTypeFeature(
        "a",
        [
            np.array([10, 11], dtype=np.uint64),
            np.array([], dtype=np.uint64),
            np.array([10, 10, 12], dtype=np.uint64),
        ],
    )
    b = IDTypeFeature(
        "b",
        [
            np.array([10], dtype=np.uint64),  # same raw id as slot a: different row!
            np.array([20], dtype=np.uint64),
            np.array([], dtype=np.uint64),
        ],
    )
    labels = [Label(np.ones((3, 1), dtype=np.float32))]
    return PersiaBatch([a, b], labels=labels, requires_grad=requires_grad)


def _expected_row(schema, slot, raw_id, dim=4):
    cfg = schema.get_slot(slot)
    sign = int(
        hashing.apply_prefix(
            np.array([raw_id], dtype=np.uint64), cfg.index_prefix, schema.feature_spacing
        )[0]
    )
    return torch.from_numpy(row_init(sign, dim, -0.5, 0.5))


def test_forward_sum_values():
    schema = _schema()
    eng = _engine(schema)
    tb = eng.process_batch(_batch())
    assert len(tb.payloads) == 2
    pa, pb = tb.payloads
    assert pa.name == "a" and pb.name == "b"
    assert pa.sum_tensor.shape == (3, 4)
    r10 = _expected_row(schema, "a", 10)
    r11 = _expected_row(schema, "a", 11)
    r12 = _expected_row(schema, "a", 12)
    assert torch.allclose(pa.sum_tensor[0].float(), (r10 + r11), atol=1e-2)
    assert torch.all(pa.sum_tensor[1] == 0)
    assert torch.allclose(pa.sum_tensor[2].float(), (2 * r10 + r12), atol=1e-2)
    # slot b sees a *different* row for raw id 10 (feature-group prefix)
    rb10 = _expected_row(schema, "b", 10)
    assert not torch.allclose(rb10, r10)
    assert torch.allclose(pb.sum_tensor[0].float(), rb10, atol=1e-2)


def test_forward_raw_contract():
    schema = _schema(sum_a=False)
    eng = _engine(schema)
    tb = eng.process_batch(_batch())
    pa = tb.payloads[0]
    assert pa.is_raw
    # distinct rows: ids {10, 11, 12} -> 3 distinct + padding
    assert pa.raw_distinct.shape == (4, 4)
    assert torch.all(pa.raw_distinct[0] == 0)
    assert pa.raw_index.shape == (9,)
    idx = pa.raw_index.view(3, 3)
    assert torch.all(idx[1] == 0)  # empty sample
    # sample 2 has [10, 10, 12] -> same distinct idx twice then another
    assert idx[2][0] == idx[2][1]
    assert idx[2][2] != idx[2][0] and idx[2][2] != 0
    assert pa.raw_sample_id_num.tolist() == [2, 0, 3]
    # row content matches seeded init
    r10 = _expected_row(schema, "a", 10)
    d_idx = int(idx[2][0])
    assert torch.allclose(pa.raw_distinct[d_idx].float(), r10, atol=1e-2)


def test_backward_sum_updates_store():
    schema = _schema()
    eng = _engine(schema)
    tb = eng.process_batch(_batch())
    r10 = _expected_row(schema, "a", 10)
    g = torch.zeros(3, 4, dtype=torch.float16)
    g[0] = 1.0  # sample 0 contains ids 10, 11
    g[2] = 0.5  # sample 2 contains 10 x2 + 12
    eng.apply_gradients(tb, {"a": g, "b": None})
    tb2 = eng.process_batch(_batch())
    # id 10 grad = 1.0 (sample0) + 0.5*2 (sample2, multiplicity 2) = 2.0
    expected = r10 - 0.1 * 2.0
    got = tb2.payloads[0]
    # extract row for id 10 from sum of sample 1... use a fresh lookup:
    eng2_row = eng.stores[4].lookup(
        torch.from_numpy(
            hashing.splitmix64(
                hashing.apply_prefix(
                    np.array([10], dtype=np.uint64),
                    schema.slots["a"].index_prefix,
                    schema.feature_spacing,
                )
            ).view(np.int64)
        ),
        train=False,
    )
    assert torch.allclose(eng2_row[0], expected, atol=1e-2)


def test_nan_gradients_skipped():
    schema = _schema()
    eng = _engine(schema)
    tb = eng.process_batch(_batch())
    before = eng.stores[4].arena.clone()
    g = torch.full((3, 4), float("nan"), dtype=torch.float16)
    eng.apply_gradients(tb, {"a": g, "b": None})
    assert torch.allclose(eng.stores[4].arena, before, equal_nan=True)
    assert eng.nan_grad_batches == 1


def test_payloads_lazy_materialization_and_order():
    """PersiaTrainingBatch.payloads is built lazily (the flagship loop never
    touches it); on access it must materialize exactly once, in the original
    id_type_features order, and stay stable across reads."""
    schema = _schema()
    eng = _engine(schema)
    tb = eng.process_batch(_batch())
    # simulate deferred construction state (CPU path builds eagerly; the
    # property contract must hold regardless)
    first = tb.payloads
    names = [p.name for p in first]
    assert names == ["a", "b"]
    again = tb.payloads
    assert again is first  # stable list identity, no rebuild
    assert tb.training_embeddings()[0].shape[0] == tb.batch_size
