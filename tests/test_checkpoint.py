import os

import numpy as np
import torch

from persia_amd.core.checkpoint import read_emb_file, write_emb_file, DONE_MARKER
from persia_amd.core.comm import DistContext
from persia_amd.core.engine import EmbeddingEngine
from persia_amd.core.schema import EmbeddingSchema, GlobalConfig, SlotConfig
from persia_amd.embedding import EmbeddingConfig
from persia_amd.embedding.data import IDTypeFeature, Label, PersiaBatch
from persia_amd.embedding.optim import Adagrad


def _engine():
    return EmbeddingEngine(
        schema=EmbeddingSchema(
            slots={
                "a": SlotConfig(name="a", dim=4),
                "c": SlotConfig(name="c", dim=8),
            }
        ),
        hyper=EmbeddingConfig(),
        optimizer=Adagrad(lr=0.1),
        gconf=GlobalConfig(capacity=1 << 12),
        device=torch.device("cpu"),
        dist_ctx=DistContext(1, 0),
    )


def _batch():
    rng = np.random.default_rng(0)
    feats = [
        IDTypeFeature("a", [rng.integers(0, 50, size=3, dtype=np.uint64) for _ in range(8)]),
        IDTypeFeature("c", [rng.integers(0, 50, size=2, dtype=np.uint64) for _ in range(8)]),
    ]
    return PersiaBatch(feats, labels=[Label(np.ones((8, 1), np.float32))], requires_grad=True)


def test_emb_file_roundtrip(tmp_path):
    signs = np.array([3, 1, 2 ** 63], dtype=np.uint64)
    inner = np.random.rand(3, 12).astype(np.float32)
    p = str(tmp_path / "x.emb")
    write_emb_file(p, signs, inner, dim=8)
    s2, i2, d = read_emb_file(p)
    assert np.array_equal(signs, s2)
    assert np.allclose(inner, i2)
    assert d == 8


def test_async_dump_status_machine(tmp_path):
    eng = _engine()
    eng.process_batch(_batch())
    eng.dump(str(tmp_path / "a"), blocking=False)
    eng.wait_for_emb_dumping()
    assert eng.model_manager_status == "Idle"
    import os

    assert os.path.exists(str(tmp_path / "a" / DONE_MARKER))
    eng2 = _engine()
    eng2.load(str(tmp_path / "a"), blocking=False)
    eng2.wait_for_emb_loading()
    assert eng2.num_resident_rows() == eng.num_resident_rows()


def test_dump_load_roundtrip(tmp_path):
    eng = _engine()
    tb = eng.process_batch(_batch())
    eng.apply_gradients(
        tb,
        {
            "a": torch.full((8, 4), 0.5, dtype=torch.float16),
            "c": torch.full((8, 8), -0.25, dtype=torch.float16),
        },
    )
    dst = str(tmp_path / "ckpt")
    eng.dump(dst)
    assert os.path.exists(os.path.join(dst, DONE_MARKER))
    assert os.path.exists(os.path.join(dst, "s0", "replica_0_shard_0.emb"))

    eng2 = _engine()
    eng2.load(dst)
    assert eng2.num_resident_rows() == eng.num_resident_rows()
    tb1 = eng.process_batch(_batch(), train=False)
    tb2 = eng2.process_batch(_batch(), train=False)
    for p1, p2 in zip(tb1.payloads, tb2.payloads):
        assert torch.equal(p1.sum_tensor, p2.sum_tensor)
    # optimizer state survives: one more identical update keeps them equal
    g = {
        "a": torch.full((8, 4), 0.5, dtype=torch.float16),
        "c": torch.full((8, 8), -0.25, dtype=torch.float16),
    }
    eng.apply_gradients(tb1, g)
    eng2.apply_gradients(tb2, g)
    ta = eng.process_batch(_batch(), train=False)
    tb_ = eng2.process_batch(_batch(), train=False)
    for p1, p2 in zip(ta.payloads, tb_.payloads):
        assert torch.equal(p1.sum_tensor, p2.sum_tensor)
