"""Reproducible-mode reorder buffer (reference PerisaDataOrderManager)."""
import numpy as np
import torch

from persia_amd.core.comm import DistContext
from persia_amd.core.engine import EmbeddingEngine, ForwardPipeline
from persia_amd.core.schema import EmbeddingSchema, GlobalConfig, SlotConfig
from persia_amd.embedding import EmbeddingConfig
from persia_amd.embedding.data import IDTypeFeature, Label, PersiaBatch
from persia_amd.embedding.optim import SGD


def _engine():
    return EmbeddingEngine(
        schema=EmbeddingSchema(slots={"a": SlotConfig(name="a", dim=4)}),
        hyper=EmbeddingConfig(),
        optimizer=SGD(lr=0.1),
        gconf=GlobalConfig(capacity=1 << 12),
        device=torch.device("cpu"),
        dist_ctx=DistContext(1, 0),
    )


def _batch(bid):
    b = PersiaBatch(
        [IDTypeFeature("a", [np.array([bid], dtype=np.uint64) for _ in range(2)])],
        labels=[Label(np.ones((2, 1), np.float32))],
        requires_grad=True,
    )
    b.batch_id = bid
    return b


def test_out_of_order_batches_are_reordered():
    pipeline = ForwardPipeline(_engine(), staleness=8, reorder=True)
    order = [3, 0, 2, 1, 4]
    for bid in order:
        pipeline.put(_batch(bid))
    pipeline.finish()
    got = []
    while True:
        tb = pipeline.get(timeout=30)
        if tb is None:
            break
        got.append(tb.batch_id)
        pipeline.release_permit()
    pipeline.stop()
    assert got == [0, 1, 2, 3, 4]


def test_in_order_without_reorder():
    pipeline = ForwardPipeline(_engine(), staleness=8, reorder=False)
    for bid in (2, 0, 1):
        pipeline.put(_batch(bid))
    pipeline.finish()
    got = []
    while True:
        tb = pipeline.get(timeout=30)
        if tb is None:
            break
        got.append(tb.batch_id)
        pipeline.release_permit()
    pipeline.stop()
    assert got == [2, 0, 1]  # arrival order preserved
