"""Synthetic adult-income-shaped dataset (there is no network access for the
real UCI data; the generator mirrors the schema of the reference example:
5 continuous features, 8 categorical slots of dim 8, binary label with a
learnable dependence on both)."""
from typing import List

import numpy as np

SLOT_NAMES = [
    "workclass",
    "education",
    "marital_status",
    "occupation",
    "relationship",
    "race",
    "gender",
    "native_country",
]
SLOT_CARDINALITY = [9, 16, 7, 15, 6, 5, 2, 42]
NUM_DENSE = 5


def make_dataset(n_samples: int = 26048, seed: int = 7):
    """-> (dense f32 [n,5], ids u64 [n,8], labels f32 [n,1])"""
    rng = np.random.default_rng(seed)
    ids = np.stack(
        [rng.integers(0, c, size=n_samples, dtype=np.uint64) for c in SLOT_CARDINALITY],
        axis=1,
    )
    dense = rng.normal(size=(n_samples, NUM_DENSE)).astype(np.float32)
    # logit = dense weights + per-(slot,id) effect
    w = rng.normal(size=NUM_DENSE) * 0.5
    effects = [rng.normal(size=c) for c in SLOT_CARDINALITY]
    logit = dense @ w
    for j, eff in enumerate(effects):
        logit += eff[ids[:, j].astype(np.int64)]
    prob = 1.0 / (1.0 + np.exp(-logit))
    labels = (rng.random(n_samples) < prob).astype(np.float32).reshape(-1, 1)
    return dense, ids, labels


def make_dataloader(dense, ids, labels, batch_size=128, skip_last_batch=False):
    from persia_amd.embedding.data import IDTypeFeatureWithSingleID, Label, NonIDTypeFeature

    n = len(labels)
    n_batches = (n - 1) // batch_size + 1
    if skip_last_batch:
        n_batches -= 1
    for b in range(n_batches):
        s, e = b * batch_size, min((b + 1) * batch_size, n)
        feats = [
            IDTypeFeatureWithSingleID(name, np.ascontiguousarray(ids[s:e, j]))
            for j, name in enumerate(SLOT_NAMES)
        ]
        yield (
            NonIDTypeFeature(dense[s:e]),
            feats,
            Label(labels[s:e]),
        )
