import numpy as np
import pytest

from persia_amd.embedding.data import (
    IDTypeFeature,
    IDTypeFeatureWithSingleID,
    Label,
    NonIDTypeFeature,
    PersiaBatch,
)


def _lil(batch, max_ids=5, seed=0):
    rng = np.random.default_rng(seed)
    return [
        rng.integers(0, 1000, size=rng.integers(0, max_ids), dtype=np.uint64)
        for _ in range(batch)
    ]


def test_id_type_feature_dtype_check():
    with pytest.raises(AssertionError):
        IDTypeFeature("f", [np.array([1, 2], dtype=np.int64)])
    with pytest.raises(AssertionError):
        IDTypeFeature("f", [np.array([[1]], dtype=np.uint64)])


def test_requires_grad_without_label_raises():
    f = IDTypeFeature("f", _lil(4))
    with pytest.raises(AssertionError):
        PersiaBatch([f], requires_grad=True)


def test_batch_size_mismatch_raises():
    f1 = IDTypeFeature("f1", _lil(4))
    f2 = IDTypeFeature("f2", _lil(5))
    with pytest.raises(AssertionError):
        PersiaBatch([f1, f2], requires_grad=False)


def test_flatten_csr():
    data = [
        np.array([], dtype=np.uint64),
        np.array([10001], dtype=np.uint64),
        np.array([7, 8, 9], dtype=np.uint64),
    ]
    f = IDTypeFeature("f", data)
    values, offsets = f.flatten()
    assert np.array_equal(offsets, [0, 0, 1, 4])
    assert np.array_equal(values, [10001, 7, 8, 9])
    single = IDTypeFeatureWithSingleID("s", np.array([1, 2, 3], dtype=np.uint64))
    v2, o2 = single.flatten()
    assert np.array_equal(v2, [1, 2, 3])
    assert np.array_equal(o2, [0, 1, 2, 3])


def test_serialization_roundtrip():
    batch = PersiaBatch(
        [IDTypeFeature("f", _lil(4, seed=3))],
        non_id_type_features=[NonIDTypeFeature(np.random.rand(4, 3).astype(np.float32))],
        labels=[Label(np.ones((4, 1), dtype=np.float32))],
        requires_grad=True,
        meta=b"hello",
    )
    batch.batch_id = 42
    again = PersiaBatch.from_bytes(batch.to_bytes())
    assert again.batch_size == 4
    assert again.requires_grad
    assert again.batch_id == 42
    assert again.meta == b"hello"
    assert len(again.id_type_features) == 1
    a, b = batch.id_type_features[0], again.id_type_features[0]
    assert a.name == b.name
    assert np.array_equal(a.values, b.values)
    assert np.array_equal(a.offsets, b.offsets)
    assert np.allclose(batch.non_id_type_features[0].data, again.non_id_type_features[0].data)
    assert np.allclose(batch.labels[0].data, again.labels[0].data)


def test_max_batch_size():
    with pytest.raises(AssertionError):
        PersiaBatch(
            [IDTypeFeatureWithSingleID("s", np.zeros(70000, dtype=np.uint64))],
            requires_grad=False,
        )
