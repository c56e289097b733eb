"""InferCtx, incremental updates, DataCtx->StreamingDataset dataflow (both
in-process and over the TCP batch queue)."""
import threading
import time

import numpy as np
import torch

from persia_amd.core.incremental import IncrementalUpdateDumper, IncrementalUpdateLoader
from persia_amd.core.schema import EmbeddingSchema, GlobalConfig, SlotConfig
from persia_amd.ctx import DataCtx, InferCtx, TrainCtx
from persia_amd.data import DataLoader, StreamingDataset
from persia_amd.embedding import EmbeddingConfig
from persia_amd.embedding.data import IDTypeFeature, Label, NonIDTypeFeature, PersiaBatch
from persia_amd.embedding.optim import SGD
from persia_amd.helper import PersiaServiceCtx
from persia_amd.models import CTRModel
from persia_amd.utils import find_free_port


def _schema():
    return EmbeddingSchema(
        slots={f"s{i}": SlotConfig(name=f"s{i}", dim=8) for i in range(2)}
    )


def _batch(requires_grad=True, seed=0, B=8):
    rng = np.random.default_rng(seed)
    feats = [
        IDTypeFeature(
            f"s{i}", [rng.integers(0, 100, size=2, dtype=np.uint64) for _ in range(B)]
        )
        for i in range(2)
    ]
    return PersiaBatch(
        feats,
        non_id_type_features=[NonIDTypeFeature(rng.normal(size=(B, 5)).astype(np.float32))],
        labels=[Label(np.ones((B, 1), np.float32))] if requires_grad else None,
        requires_grad=requires_grad,
    )


def test_infer_ctx_from_bytes(tmp_path):
    # train a few steps, dump, load into a fresh InferCtx, infer from bytes
    model = CTRModel(num_dense=5, sparse_input_dim=16)
    with TrainCtx(
        model=model,
        embedding_optimizer=SGD(lr=0.1),
        dense_optimizer=torch.optim.SGD(model.parameters(), lr=0.05),
        embedding_schema=_schema(),
        global_config=GlobalConfig(capacity=1 << 12),
        mixed_precision=False,
    ) as ctx:
        for step in range(3):
            tb = ctx.engine.process_batch(_batch(seed=step))
            pred, labels = ctx.forward(tb)
            loss = torch.nn.functional.binary_cross_entropy(
                pred.squeeze(1), labels[0].squeeze(1)
            )
            ctx.backward(loss)
        ctx.dump_checkpoint(str(tmp_path / "ckpt"))
        resident = ctx.engine.num_resident_rows()
    assert resident > 0

    model2 = CTRModel(num_dense=5, sparse_input_dim=16)
    ictx = InferCtx(
        model=model2,
        embedding_schema=_schema(),
        global_config=GlobalConfig(capacity=1 << 12),
    )
    ictx.load_checkpoint(str(tmp_path / "ckpt"))
    assert ictx.engine.num_resident_rows() == resident
    model2.eval()
    payload = _batch(requires_grad=False, seed=99).to_bytes()
    tb = ictx.get_embedding_from_bytes(payload)
    pred, _ = ictx.forward(tb)
    assert pred.shape == (8, 1)
    assert torch.isfinite(pred).all()
    # infer mode never inserts
    before = ictx.engine.num_resident_rows()
    ictx.get_embedding_from_data(_batch(requires_grad=False, seed=123))
    assert ictx.engine.num_resident_rows() == before


def test_incremental_update_roundtrip(tmp_path):
    from persia_amd.core.comm import DistContext
    from persia_amd.core.engine import EmbeddingEngine

    train_eng = EmbeddingEngine(
        schema=_schema(), hyper=EmbeddingConfig(), optimizer=SGD(lr=0.1),
        gconf=GlobalConfig(capacity=1 << 12), device=torch.device("cpu"),
        dist_ctx=DistContext(1, 0),
    )
    tb = train_eng.process_batch(_batch(seed=1))
    train_eng.apply_gradients(
        tb, {"s0": torch.ones(8, 8, dtype=torch.float16),
             "s1": torch.ones(8, 8, dtype=torch.float16)},
    )
    dumper = IncrementalUpdateDumper(train_eng, str(tmp_path), buffer_size=10 ** 6)
    signs, _ = train_eng.stores[8].export_rows()
    dumper.record(signs)
    assert dumper.flush() is not None

    infer_eng = EmbeddingEngine(
        schema=_schema(), hyper=EmbeddingConfig(), optimizer=SGD(lr=0.1),
        gconf=GlobalConfig(capacity=1 << 12, job_type="infer"),
        device=torch.device("cpu"), dist_ctx=DistContext(1, 0),
    )
    loader = IncrementalUpdateLoader(infer_eng, str(tmp_path))
    loaded = loader.scan_once()
    assert loaded == len(signs)
    t1 = train_eng.process_batch(_batch(seed=1), train=False)
    t2 = infer_eng.process_batch(_batch(seed=1), train=False)
    for p1, p2 in zip(t1.payloads, t2.payloads):
        assert torch.equal(p1.sum_tensor, p2.sum_tensor)


def test_dataflow_in_process():
    with PersiaServiceCtx():
        dctx = DataCtx()
        for i in range(3):
            dctx.send_data(_batch(seed=i))
        ds = StreamingDataset(buffer_size=10)
        got = []
        it = ds.batches()
        for _ in range(3):
            got.append(next(it))
        assert [b.batch_id for b in got] == [0, 1, 2]


def test_dataflow_tcp():
    from persia_amd.core.queue import BatchQueueClient, BatchQueueServer, PersiaBatchDataChannel

    channel = PersiaBatchDataChannel(10)
    server = BatchQueueServer(0, channel)
    client = BatchQueueClient("127.0.0.1", server.port)
    sent = _batch(seed=7)
    sent.batch_id = 5
    client.send(sent)
    receiver = channel.get_receiver()
    got = receiver.recv(timeout=10)
    assert got is not None
    assert got.batch_id == 5
    assert np.array_equal(
        got.id_type_features[0].values, sent.id_type_features[0].values
    )
    client.close()
    server.close()
