import numpy as np

from persia_amd.core import hashing
from persia_amd.core.schema import EmbeddingSchema, HashStackConfig, SlotConfig


def test_splitmix64_bijective_roundtrip():
    rng = np.random.default_rng(0)
    x = rng.integers(0, 2 ** 64, size=10000, dtype=np.uint64)
    h = hashing.splitmix64(x)
    assert np.array_equal(hashing.splitmix64_inv(h), x)
    # includes edge values
    edges = np.array([0, 1, 2 ** 63, 2 ** 64 - 1], dtype=np.uint64)
    assert np.array_equal(hashing.splitmix64_inv(hashing.splitmix64(edges)), edges)


def test_owner_partition_monotone_and_balanced():
    rng = np.random.default_rng(1)
    x = rng.integers(0, 2 ** 64, size=200000, dtype=np.uint64)
    h = np.sort(hashing.splitmix64(x))
    for w in (1, 2, 3, 4, 8):
        owner = hashing.owner_of(h, w)
        assert owner.min() >= 0 and owner.max() < w
        # monotone in sorted order -> contiguous per owner
        assert np.all(np.diff(owner) >= 0)
        if w > 1:
            counts = np.bincount(owner, minlength=w)
            assert counts.min() > 0.8 * len(h) / w


def test_apply_prefix_disjoint_groups():
    schema = EmbeddingSchema(
        slots={
            "a": SlotConfig(name="a", dim=8),
            "b": SlotConfig(name="b", dim=8),
        },
        feature_index_prefix_bit=8,
    )
    spacing = schema.feature_spacing
    ids = np.arange(100, dtype=np.uint64)
    pa = hashing.apply_prefix(ids, schema.slots["a"].index_prefix, spacing)
    pb = hashing.apply_prefix(ids, schema.slots["b"].index_prefix, spacing)
    assert len(np.intersect1d(pa, pb)) == 0
    # prefix occupies the top bits: group index recoverable
    assert np.all((pa >> np.uint64(56)) == 1)
    assert np.all((pb >> np.uint64(56)) == 2)
    # reference assignment formula (config lib.rs:600-650)
    assert schema.slots["a"].index_prefix == 1 << 56
    assert schema.slots["b"].index_prefix == 2 << 56


def test_hash_stack_buckets():
    ids = np.arange(1000, dtype=np.uint64)
    out = hashing.hash_stack(ids, rounds=3, embedding_size=100)
    assert out.shape == (3, 1000)
    for r in range(3):
        assert out[r].min() >= r * 100
        assert out[r].max() < (r + 1) * 100
    # deterministic
    again = hashing.hash_stack(ids, rounds=3, embedding_size=100)
    assert np.array_equal(out, again)
