"""HTTP inference server (persia_amd.serving): the runnable analog of the
reference's TorchServe integration (resources/proto/inference.proto: Ping +
Predictions bytes-in/bytes-out) — checkpoint -> handler -> HTTP byte wire
-> f32 predictions identical to the in-process handler."""
import numpy as np
import pytest
import torch

from persia_amd.core.schema import EmbeddingSchema, GlobalConfig, SlotConfig
from persia_amd.ctx import TrainCtx
from persia_amd.embedding.data import (
    IDTypeFeature,
    Label,
    NonIDTypeFeature,
    PersiaBatch,
)
from persia_amd.embedding.optim import SGD
from persia_amd.models import CTRModel


def _schema():
    return EmbeddingSchema(
        slots={f"s{i}": SlotConfig(name=f"s{i}", dim=8) for i in range(2)}
    )


def _batch(requires_grad=True, seed=0, B=8):
    rng = np.random.default_rng(seed)
    feats = [
        IDTypeFeature(
            f"s{i}",
            [rng.integers(0, 100, size=2, dtype=np.uint64) for _ in range(B)],
        )
        for i in range(2)
    ]
    return PersiaBatch(
        feats,
        non_id_type_features=[
            NonIDTypeFeature(rng.normal(size=(B, 5)).astype(np.float32))
        ],
        labels=[Label(np.ones((B, 1), np.float32))] if requires_grad else None,
        requires_grad=requires_grad,
    )


def test_http_inference_roundtrip(tmp_path):
    pytest.importorskip("fastapi")
    from fastapi.testclient import TestClient

    from persia_amd.serving import PersiaHandler, create_app

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
            ctx.backward(
                torch.nn.functional.binary_cross_entropy(
                    pred.squeeze(1), labels[0].squeeze(1)
                )
            )
        ctx.dump_checkpoint(str(tmp_path / "ckpt"))

    model2 = CTRModel(num_dense=5, sparse_input_dim=16)
    model2.load_state_dict(model.state_dict())
    handler = PersiaHandler(
        model2, embedding_schema=_schema(),
        checkpoint_dir=str(tmp_path / "ckpt"),
    )
    wire = _batch(requires_grad=False, seed=7).to_bytes()
    want = np.asarray(handler.handle(wire), dtype=np.float32)

    with TestClient(create_app(handler)) as c:
        assert c.get("/ping").json() == {"health": "healthy"}
        r = c.post("/predictions/persia", content=wire)
        assert r.status_code == 200
        got = np.frombuffer(r.content, dtype="<f4")
        np.testing.assert_allclose(got, want, rtol=1e-6)
        # trained rows loaded: predictions are not the zeros-on-miss output
        assert np.std(got) > 0
        assert c.post("/predictions/nope", content=wire).status_code == 404
        assert c.post("/predictions/persia", content=b"junk").status_code == 400
